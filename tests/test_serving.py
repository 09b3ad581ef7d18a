"""ProbeServer: HTTPS metrics endpoint with cert hot-reload and authn filter.

Mirrors the reference's metrics-server behaviors (cmd/main.go:122-199):
certwatcher reload, WithAuthenticationAndAuthorization filter, TLS>=1.2,
unauthenticated healthz/readyz.
"""
from __future__ import annotations

import ssl
import subprocess
import time

import httpx
import pytest

from inferno_amd.controller.serving import MetricsAuth, ProbeServer


def _make_cert(tmp_path, cn="localhost", name="srv"):
    key = tmp_path / f"{name}-tls.key"
    crt = tmp_path / f"{name}-tls.crt"
    subprocess.run(
        [
            "openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
            "-keyout", str(key), "-out", str(crt), "-days", "1",
            "-subj", f"/CN={cn}", "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1",
        ],
        check=True, capture_output=True,
    )
    return crt, key


@pytest.fixture
def plain_server():
    state = {"ready": False}
    srv = ProbeServer(0, state, bind="127.0.0.1")
    yield srv, state
    srv.shutdown()


class TestPlainHTTP:
    def test_probes_and_metrics(self, plain_server):
        srv, state = plain_server
        base = f"http://127.0.0.1:{srv.port}"
        assert httpx.get(f"{base}/healthz").status_code == 200
        assert httpx.get(f"{base}/readyz").status_code == 503
        state["ready"] = True
        assert httpx.get(f"{base}/readyz").status_code == 200
        r = httpx.get(f"{base}/metrics")
        assert r.status_code == 200
        assert "python_info" in r.text or "inferno" in r.text

    def test_404(self, plain_server):
        srv, _ = plain_server
        assert httpx.get(f"http://127.0.0.1:{srv.port}/nope").status_code == 404


class TestTLS:
    def test_https_serving_and_min_version(self, tmp_path):
        crt, key = _make_cert(tmp_path)
        cert_dir = tmp_path
        (cert_dir / "tls.crt").write_bytes(crt.read_bytes())
        (cert_dir / "tls.key").write_bytes(key.read_bytes())
        srv = ProbeServer(0, {"ready": True}, cert_dir=str(cert_dir), bind="127.0.0.1")
        try:
            assert srv.tls
            ctx = ssl.create_default_context(cafile=str(cert_dir / "tls.crt"))
            ctx.check_hostname = False
            r = httpx.get(f"https://127.0.0.1:{srv.port}/healthz", verify=ctx)
            assert r.status_code == 200
            # plain HTTP against the TLS port must fail
            with pytest.raises(httpx.HTTPError):
                httpx.get(f"http://127.0.0.1:{srv.port}/healthz", timeout=2)
        finally:
            srv.shutdown()

    def test_cert_hot_reload(self, tmp_path):
        crt, key = _make_cert(tmp_path, cn="old")
        (tmp_path / "tls.crt").write_bytes(crt.read_bytes())
        (tmp_path / "tls.key").write_bytes(key.read_bytes())
        srv = ProbeServer(0, {"ready": True}, cert_dir=str(tmp_path), bind="127.0.0.1")
        srv.watcher.poll_seconds = 0.05
        try:
            def serial():
                ctx = ssl.create_default_context()
                ctx.check_hostname = False
                ctx.verify_mode = ssl.CERT_NONE
                import socket

                with socket.create_connection(("127.0.0.1", srv.port), timeout=3) as s:
                    with ctx.wrap_socket(s) as tls:
                        return tls.getpeercert(binary_form=True)

            before = serial()
            crt2, key2 = _make_cert(tmp_path, cn="new", name="n")
            (tmp_path / "tls.crt").write_bytes(crt2.read_bytes())
            (tmp_path / "tls.key").write_bytes(key2.read_bytes())
            deadline = time.time() + 5
            after = before
            while time.time() < deadline and after == before:
                time.sleep(0.1)
                after = serial()
            assert after != before, "certificate was not hot-reloaded"
        finally:
            srv.shutdown()


class TestAuth:
    def test_static_token(self, tmp_path):
        tok = tmp_path / "token"
        tok.write_text("s3cret\n")
        srv = ProbeServer(0, {"ready": True},
                          auth=MetricsAuth(token_file=str(tok)), bind="127.0.0.1")
        try:
            base = f"http://127.0.0.1:{srv.port}"
            assert httpx.get(f"{base}/metrics").status_code == 401
            assert httpx.get(f"{base}/metrics",
                             headers={"Authorization": "Bearer wrong"}).status_code == 401
            assert httpx.get(f"{base}/metrics",
                             headers={"Authorization": "Bearer s3cret"}).status_code == 200
            # probes stay unauthenticated (controller-runtime behavior)
            assert httpx.get(f"{base}/healthz").status_code == 200
        finally:
            srv.shutdown()

    def test_tokenreview_delegation(self):
        """Bearer tokens delegated to the TokenReview API like
        filters.WithAuthenticationAndAuthorization (cmd/main.go:157-169)."""
        import json as _json

        def handler(req):
            assert req.url.path == "/apis/authentication.k8s.io/v1/tokenreviews"
            body = _json.loads(req.content)
            ok = body["spec"]["token"] == "good"
            return httpx.Response(201, json={"status": {"authenticated": ok}})

        class FakeKube:
            _client = httpx.Client(base_url="https://kube.test",
                                   transport=httpx.MockTransport(handler))

        auth = MetricsAuth(kube=FakeKube())
        assert auth.enabled
        assert auth.check("Bearer good") is True
        assert auth.check("Bearer bad") is False
        assert auth.check("") is False


class TestProbeOnlyListener:
    def test_split_port_layout(self):
        """Probe-only listener (controller-runtime's 8081/8443 split): serves
        healthz/readyz but NOT /metrics."""
        srv = ProbeServer(0, {"ready": True}, bind="127.0.0.1",
                          expose_metrics=False)
        try:
            base = f"http://127.0.0.1:{srv.port}"
            assert httpx.get(f"{base}/healthz").status_code == 200
            assert httpx.get(f"{base}/readyz").status_code == 200
            assert httpx.get(f"{base}/metrics").status_code == 404
        finally:
            srv.shutdown()
