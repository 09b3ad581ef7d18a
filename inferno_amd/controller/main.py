"""Controller process entrypoint.

Equivalent of cmd/main.go:62-275: flag/env parsing, JSON logging, metrics
endpoint, health probes, lease-based leader election, GPU device init/health
probe (MI355X addition), then the reconcile loop at GLOBAL_OPT_INTERVAL.

Run: python -m inferno_amd.controller.main [--metrics-port 8443]
"""
from __future__ import annotations

import argparse
import os
import signal
import socket
import threading
import time

# one HW queue per sweep bucket stream (ROCm defaults to 4; set before HIP
# runtime init so the overlapped bucket launches don't serialize)
os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")

from ..utils.logging import init_logger
from . import collector
from .k8s import HttpKube
from .leader import LeaderElector
from .metrics import init_metrics
from .reconciler import CONFIGMAP_NAMESPACE, Reconciler, WVA_CONFIG_CM


def gpu_health_probe() -> dict:
    """MI355X device probe used by readyz (HIP addition over the reference)."""
    try:
        import torch

        if not torch.cuda.is_available():
            return {"gpu": False, "reason": "no HIP device"}
        from ..ops.sweep import load_library

        load_library(allow_build=False)
        return {"gpu": True, "device": torch.cuda.get_device_name(0)}
    except Exception as e:  # noqa: BLE001
        return {"gpu": False, "reason": str(e)}


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--metrics-port", type=int, default=8443)
    p.add_argument("--probe-port", type=int, default=0,
                   help="serve healthz/readyz on a separate plain-HTTP port "
                        "(controller-runtime's split layout: probes on 8081, "
                        "secured metrics on 8443); 0 = same port as metrics")
    p.add_argument("--metrics-cert-dir", default=os.environ.get("METRICS_CERT_DIR", ""),
                   help="dir with tls.crt/tls.key; enables HTTPS + hot reload "
                        "(ref cmd/main.go:122-155 certwatcher)")
    p.add_argument("--metrics-client-ca", default=os.environ.get("METRICS_CLIENT_CA", ""),
                   help="CA bundle; requires verified client certs on /metrics")
    p.add_argument("--metrics-auth-token-file",
                   default=os.environ.get("METRICS_AUTH_TOKEN_FILE", ""),
                   help="static bearer token file protecting /metrics")
    p.add_argument("--metrics-auth-k8s", action="store_true",
                   default=os.environ.get("METRICS_AUTH_K8S", "") == "true",
                   help="delegate /metrics bearer tokens to the TokenReview API "
                        "(ref cmd/main.go:157-169 authn/authz filter)")
    p.add_argument("--leader-elect", action="store_true", default=True)
    p.add_argument("--no-leader-elect", dest="leader_elect", action="store_false")
    p.add_argument("--configmap-namespace", default=CONFIGMAP_NAMESPACE)
    p.add_argument("--backend", choices=["auto", "gpu", "cpu"], default="auto")
    args = p.parse_args()

    logger = init_logger()
    state = {"ready": False}

    kube = HttpKube()

    from .serving import MetricsAuth, ProbeServer

    auth = MetricsAuth(
        token_file=args.metrics_auth_token_file or None,
        kube=kube if args.metrics_auth_k8s else None,
    )
    probe_server = ProbeServer(
        args.metrics_port, state,
        cert_dir=args.metrics_cert_dir or None,
        client_ca=args.metrics_client_ca or None,
        auth=auth,
    )
    if args.probe_port:
        # plain-HTTP probe-only listener for the kubelet (no /metrics)
        ProbeServer(args.probe_port, state, expose_metrics=False)
    logger.info(
        "metrics/probe server started",
        extra={"kv": {"port": probe_server.port, "tls": probe_server.tls,
                      "auth": auth.enabled, "probe_port": args.probe_port or None}},
    )
    cm = kube.get_configmap(args.configmap_namespace, WVA_CONFIG_CM) or {}
    prom_cfg = collector.prometheus_config_from_env(cm)
    prom = collector.PrometheusClient(**prom_cfg)
    # startup connectivity validation with backoff (reference: "up" query)
    try:
        collector.validate_prometheus_api(prom)
    except Exception as e:  # noqa: BLE001
        logger.error("Prometheus validation failed after retries", extra={"kv": {"err": str(e)}})
        raise SystemExit(1)

    emitter = init_metrics()
    reconciler = Reconciler(
        kube, prom, emitter, backend=args.backend,
        configmap_namespace=args.configmap_namespace,
    )

    logger.info("starting controller", extra={"kv": gpu_health_probe()})

    elector = None
    if args.leader_elect:
        identity = f"{socket.gethostname()}_{os.getpid()}"
        # LeaderElectionID mirrors the reference (cmd/main.go:207)
        elector = LeaderElector(kube, "72dd1cf1.llm-d.ai", args.configmap_namespace, identity)

    stop = threading.Event()
    wake = threading.Event()  # VA-create events trigger an immediate tick
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())

    def watch_loop():
        """Create-only VA watch (the reference's event filter: Create -> true,
        Update/Delete/Generic -> false, controller.go:456-487). Tracks the
        last seen resourceVersion across the periodic reconnects so old
        events are not replayed (which would fire spurious wake-ups)."""
        rv = ""
        while not stop.is_set():
            try:
                for etype, va in kube.watch_events(resource_version=rv,
                                                   timeout_seconds=55):
                    if va.resourceVersion:
                        rv = va.resourceVersion
                    if etype == "ADDED":
                        wake.set()
                    if stop.is_set():
                        return
            except Exception:
                stop.wait(5.0)

    def cm_watch_loop():
        """Watch the controller ConfigMaps; any change triggers a tick
        (controller.go:456-487 Watches(ConfigMap))."""
        from .reconciler import ACCELERATOR_CM, SERVICE_CLASS_CM, WVA_CONFIG_CM

        names = {ACCELERATOR_CM, SERVICE_CLASS_CM, WVA_CONFIG_CM}
        rv = ""
        while not stop.is_set():
            try:
                for _etype, _name, ev_rv in kube.watch_configmap_events(
                    args.configmap_namespace, names, timeout_seconds=55,
                    resource_version=rv,
                ):
                    if ev_rv:
                        rv = ev_rv
                    wake.set()
                    if stop.is_set():
                        return
            except Exception:
                stop.wait(5.0)

    threading.Thread(target=watch_loop, daemon=True).start()
    threading.Thread(target=cm_watch_loop, daemon=True).start()

    state["ready"] = True
    while not stop.is_set():
        if elector is not None and not elector.try_acquire():
            time.sleep(2.0)
            continue
        wake.clear()
        try:
            result = reconciler.reconcile()
        except Exception as e:  # noqa: BLE001 - last-resort guard: a tick
            # must never kill the process (transient API/Prometheus/network
            # failures degrade to an errored tick + requeue)
            logger.error("reconcile tick failed", extra={"kv": {"err": str(e)}})
            stop.wait(5.0)
            continue
        logger.info(
            "reconcile complete",
            extra={
                "kv": {
                    "processed": result.processed,
                    "backend": result.solver_backend,
                    "duration_s": round(result.duration_seconds, 4),
                    "errors": result.errors[:5],
                }
            },
        )
        # sleep until the requeue interval elapses or a VA-create event fires
        deadline = time.monotonic() + result.requeue_after
        while not stop.is_set() and time.monotonic() < deadline:
            if wake.wait(timeout=min(1.0, max(deadline - time.monotonic(), 0.05))):
                break


if __name__ == "__main__":
    main()
