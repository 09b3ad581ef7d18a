"""Probe + metrics HTTP(S) server.

The analogue of the reference's metrics-server setup in cmd/main.go:122-199:

* HTTPS when a cert dir with ``tls.crt``/``tls.key`` is provided
  (``--metrics-cert-path`` in the reference); plain HTTP otherwise, matching
  the reference's self-signed fallback for dev.
* Certificate hot-reload: a watcher thread polls the cert/key mtimes and
  atomically swaps the SSLContext (certwatcher.New, cmd/main.go:128-155) so
  cert-manager rotation needs no restart.
* authn/authz filter on /metrics (filters.WithAuthenticationAndAuthorization,
  cmd/main.go:157-169): a bearer token checked either against a static token
  file or delegated to the Kubernetes TokenReview API, and/or a verified
  client certificate when a client CA is configured. healthz/readyz stay
  unauthenticated (same as controller-runtime's probe endpoints).
"""
from __future__ import annotations

import os
import ssl
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional


class _CertWatcher:
    """Polls cert/key files and rebuilds the SSLContext on change."""

    def __init__(self, cert_file: str, key_file: str, client_ca: Optional[str],
                 poll_seconds: float = 2.0):
        self.cert_file = cert_file
        self.key_file = key_file
        self.client_ca = client_ca
        self.poll_seconds = poll_seconds
        self._mtimes = self._stat()
        self.context = self._build()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True)

    def _stat(self):
        def m(p):
            try:
                return os.stat(p).st_mtime_ns
            except OSError:
                return 0

        return (m(self.cert_file), m(self.key_file))

    def _build(self) -> ssl.SSLContext:
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.minimum_version = ssl.TLSVersion.TLSv1_2  # ref tls.go:27
        ctx.load_cert_chain(self.cert_file, self.key_file)
        if self.client_ca:
            ctx.load_verify_locations(self.client_ca)
            ctx.verify_mode = ssl.CERT_REQUIRED
        return ctx

    def start(self):
        self._thread.start()

    def stop(self):
        self._stop.set()

    def _loop(self):
        while not self._stop.wait(self.poll_seconds):
            cur = self._stat()
            if cur != self._mtimes:
                try:
                    self.context = self._build()
                    self._mtimes = cur
                except (OSError, ssl.SSLError):
                    pass  # partial rotation: keep serving with the old cert


class MetricsAuth:
    """Bearer-token authentication for /metrics.

    ``token_file`` pins a static shared token; ``kube`` (an HttpKube) delegates
    to the TokenReview API like controller-runtime's filter does. Either being
    configured turns authentication on."""

    def __init__(self, token_file: Optional[str] = None, kube=None):
        self.token_file = token_file
        self.kube = kube

    @property
    def enabled(self) -> bool:
        return bool(self.token_file) or self.kube is not None

    def check(self, authorization_header: str) -> bool:
        if not self.enabled:
            return True
        if not authorization_header.startswith("Bearer "):
            return False
        token = authorization_header[len("Bearer "):].strip()
        if not token:
            return False
        if self.token_file:
            try:
                with open(self.token_file) as f:
                    want = f.read().strip()
            except OSError:
                return False
            if want and token == want:
                return True
            if self.kube is None:
                return False
        if self.kube is not None:
            return self._token_review(token)
        return False

    def _token_review(self, token: str) -> bool:
        client = getattr(self.kube, "_client", None)
        if client is None:
            return False
        try:
            r = client.post(
                "/apis/authentication.k8s.io/v1/tokenreviews",
                json={
                    "apiVersion": "authentication.k8s.io/v1",
                    "kind": "TokenReview",
                    "spec": {"token": token},
                },
            )
            if r.status_code not in (200, 201):
                return False
            return bool((r.json().get("status") or {}).get("authenticated"))
        except Exception:
            return False


class _TLSServer(ThreadingHTTPServer):
    daemon_threads = True
    watcher: Optional[_CertWatcher] = None

    def finish_request(self, request, client_address):
        # wrap per-connection with the *current* context so hot-reloaded
        # certs take effect without rebinding the listener. Done HERE (in
        # the per-connection worker thread, not get_request) so a slow or
        # malicious client's handshake cannot stall the accept loop; a
        # failed handshake raises into handle_error and only kills this
        # connection.
        if self.watcher is not None:
            request = self.watcher.context.wrap_socket(request, server_side=True)
        super().finish_request(request, client_address)

    def handle_error(self, request, client_address):
        pass  # per-connection TLS/parse failures are not server errors


class ProbeServer:
    """healthz/readyz + prometheus /metrics on one port, TLS-capable."""

    def __init__(
        self,
        port: int,
        state: dict,
        cert_dir: Optional[str] = None,
        client_ca: Optional[str] = None,
        auth: Optional[MetricsAuth] = None,
        bind: str = "0.0.0.0",
        expose_metrics: bool = True,
    ):
        from prometheus_client import generate_latest

        auth = auth or MetricsAuth()
        self.expose_metrics = expose_metrics
        watcher: Optional[_CertWatcher] = None
        if cert_dir:
            cert = os.path.join(cert_dir, "tls.crt")
            key = os.path.join(cert_dir, "tls.key")
            if os.path.exists(cert) and os.path.exists(key):
                watcher = _CertWatcher(cert, key, client_ca)

        class Handler(BaseHTTPRequestHandler):
            def do_GET(self):  # noqa: N802
                if self.path == "/healthz":
                    self._ok(b"ok")
                elif self.path == "/readyz":
                    if state.get("ready"):
                        self._ok(b"ok")
                    else:
                        self.send_response(503)
                        self.end_headers()
                elif self.path == "/metrics":
                    if not expose_metrics:
                        # probe-only listener (split-port layout, ref
                        # cmd/main.go: probes on 8081, metrics on 8443)
                        self.send_response(404)
                        self.end_headers()
                        return
                    if not auth.check(self.headers.get("Authorization", "")):
                        self.send_response(401)
                        self.send_header("WWW-Authenticate", "Bearer")
                        self.end_headers()
                        return
                    body = generate_latest()
                    self.send_response(200)
                    self.send_header("Content-Type", "text/plain; version=0.0.4")
                    self.end_headers()
                    self.wfile.write(body)
                else:
                    self.send_response(404)
                    self.end_headers()

            def _ok(self, body: bytes):
                self.send_response(200)
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):  # silence
                pass

        self._server = _TLSServer((bind, port), Handler)
        self._server.watcher = watcher
        self.watcher = watcher
        self.tls = watcher is not None
        self.port = self._server.server_address[1]
        if watcher is not None:
            watcher.start()
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._thread.start()

    def shutdown(self):
        if self.watcher is not None:
            self.watcher.stop()
        self._server.shutdown()
        self._server.server_close()
