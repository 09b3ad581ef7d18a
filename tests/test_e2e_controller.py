"""Full-loop e2e: the REAL controller process against the apiserver stand-in,
the vLLM emulator and a TLS Prometheus stand-in, under real HTTP load.

The executed counterpart of the reference's Kind e2e
(test/e2e/e2e_test.go:341-430 scale-out under load with Prometheus
cross-checks, :519 scale-in at idle, conditions/ownerRef/lease assertions)
— every component in its own process, wire formats end to end.
"""
from __future__ import annotations

import os
import signal
import subprocess
import sys
import threading
import time

import httpx
import pytest
import yaml

from tests.test_e2e_apiserver import CRD, NS_SYS, REPO, _apply, _free_port, _spawn

VA_NS = "llm-d-sim"
VA_NAME = "vllme-deploy"
MODEL = "default/default"


def _mk_cert(tmp_path):
    import subprocess as sp

    key = tmp_path / "tls.key"
    crt = tmp_path / "tls.crt"
    sp.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(crt), "-days", "1",
         "-subj", "/CN=promstub",
         "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    return str(crt), str(key)


@pytest.fixture(scope="module")
def world(tmp_path_factory):
    """apiserver + emulator + TLS promstub + seeded objects + controller."""
    tmp = tmp_path_factory.mktemp("e2e")
    procs = []
    try:
        # 1. apiserver stand-in
        api_proc, api_port = _spawn(
            [sys.executable, "-m", "inferno_amd.testing.kubeapi", "--port", "0",
             "--crd", CRD],
            match="kubeapi listening on",
        )
        procs.append(api_proc)
        kube = httpx.Client(base_url=f"http://127.0.0.1:{api_port}", timeout=10.0)

        # 2. vLLM emulator (fast decode so load generation is cheap)
        emu_port = _free_port()
        emu_env = {
            "PORT": str(emu_port), "MODEL_NAME": MODEL, "NAMESPACE": VA_NS,
            "DECODE_TIME": "1", "PREFILL_TIME": "1", "MAX_BATCH_SIZE": "256",
        }
        emu_proc = subprocess.Popen(
            [sys.executable, "-m", "inferno_amd.emulator.server"],
            env={**os.environ, **emu_env},
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, cwd=REPO,
        )
        procs.append(emu_proc)
        emu = f"http://127.0.0.1:{emu_port}"
        for _ in range(100):
            try:
                if httpx.get(f"{emu}/healthz", timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.2)
        else:
            raise RuntimeError("emulator did not come up")

        # 3. TLS promstub scraping the emulator (reference enforces HTTPS
        #    Prometheus, tls.go:63-68 — keep that property end to end)
        crt, key = _mk_cert(tmp)
        prom_proc, prom_port = _spawn(
            [sys.executable, "-m", "inferno_amd.testing.promstub", "--port", "0",
             "--target", emu, "--interval", "0.5",
             "--tls-cert", crt, "--tls-key", key],
            match="promstub listening on",
        )
        procs.append(prom_proc)

        # 4. seed cluster objects
        for doc in yaml.safe_load_all(
            open(os.path.join(REPO, "deploy", "configmap-accelerator-unitcost.yaml"))
        ):
            if doc:
                assert _apply(kube, doc).status_code == 201
        for doc in yaml.safe_load_all(
            open(os.path.join(REPO, "deploy", "configmap-serviceclass.yaml"))
        ):
            if doc:
                assert _apply(kube, doc).status_code == 201
        assert _apply(kube, {
            "apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {
                "name": "workload-variant-autoscaler-variantautoscaling-config",
                "namespace": NS_SYS,
            },
            "data": {"GLOBAL_OPT_INTERVAL": "2s"},
        }).status_code == 201
        dep = _apply(kube, {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": VA_NAME, "namespace": VA_NS},
            "spec": {"replicas": 1}, "status": {"replicas": 1},
        })
        assert dep.status_code == 201
        with open(os.path.join(REPO, "deploy", "examples",
                               "vllme-variantautoscaling.yaml")) as f:
            va_doc = next(d for d in yaml.safe_load_all(f)
                          if d and d["metadata"]["name"] == VA_NAME)
        va_doc["metadata"]["namespace"] = VA_NS
        assert _apply(kube, va_doc).status_code == 201

        # 5. the controller process
        metrics_port = _free_port()
        ctl_env = {
            "KUBE_API_URL": f"http://127.0.0.1:{api_port}",
            "PROMETHEUS_BASE_URL": f"https://127.0.0.1:{prom_port}",
            "PROMETHEUS_CA_CERT_PATH": crt,
            "LOG_LEVEL": "info",
        }
        ctl_proc = subprocess.Popen(
            [sys.executable, "-m", "inferno_amd.controller.main",
             "--metrics-port", str(metrics_port), "--backend", "cpu"],
            env={**os.environ, **ctl_env},
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, cwd=REPO,
        )
        procs.append(ctl_proc)
        threading.Thread(target=lambda: [None for _ in ctl_proc.stdout],
                         daemon=True).start()
        # readiness
        for _ in range(150):
            try:
                if httpx.get(f"http://127.0.0.1:{metrics_port}/readyz",
                             timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            if ctl_proc.poll() is not None:
                raise RuntimeError("controller exited early")
            time.sleep(0.2)
        else:
            raise RuntimeError("controller never became ready")

        yield {
            "kube": kube, "emu": emu,
            "metrics_port": metrics_port, "api_port": api_port,
            "prom_url": f"https://127.0.0.1:{prom_port}", "prom_ca": crt,
        }
        kube.close()
    finally:
        for p in procs:
            try:
                p.send_signal(signal.SIGTERM)
            except OSError:
                pass
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()


def _va_status(kube) -> dict:
    r = kube.get(
        f"/apis/llmd.ai/v1alpha1/namespaces/{VA_NS}/variantautoscalings/{VA_NAME}"
    )
    assert r.status_code == 200, r.text
    return r.json()


def _drive_load(emu: str, seconds: float, concurrency: int = 8) -> int:
    """Hammer the emulator's OpenAI endpoint from several threads."""
    stop = time.time() + seconds
    done = [0]
    lock = threading.Lock()

    def worker():
        with httpx.Client(timeout=30.0) as c:
            while time.time() < stop:
                try:
                    c.post(f"{emu}/v1/chat/completions", json={
                        "model": MODEL,
                        "messages": [{"role": "user", "content": "lorem " * 100}],
                        "max_tokens": 40,
                    })
                    with lock:
                        done[0] += 1
                except httpx.HTTPError:
                    pass

    threads = [threading.Thread(target=worker) for _ in range(concurrency)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    return done[0]


def _wait_for(pred, timeout_s: float, interval: float = 1.0, desc: str = ""):
    deadline = time.time() + timeout_s
    last = None
    while time.time() < deadline:
        last = pred()
        if last:
            return last
        time.sleep(interval)
    raise AssertionError(f"timed out waiting for {desc}; last={last!r}")


@pytest.mark.e2e
class TestControllerEndToEnd:
    def test_full_loop(self, world):
        kube = world["kube"]
        emu = world["emu"]

        # -- leader lease appears (e2e_suite_test.go:95-110) -------------
        def lease_held():
            r = kube.get(
                f"/apis/coordination.k8s.io/v1/namespaces/{NS_SYS}"
                "/leases/72dd1cf1.llm-d.ai"
            )
            if r.status_code != 200:
                return None
            return r.json()["spec"].get("holderIdentity") or None

        holder = _wait_for(lease_held, 30, desc="leader lease")
        assert holder

        # -- drive load; controller reconciles every 2s ------------------
        n = _drive_load(emu, seconds=12.0, concurrency=8)
        assert n > 20, f"load generator only completed {n} requests"

        # -- scale-out: desired allocation computed and written ----------
        def optimized():
            va = _va_status(kube)
            st = va.get("status", {})
            des = st.get("desiredOptimizedAlloc", {})
            conds = {c["type"]: c["status"] for c in st.get("conditions", [])}
            if (des.get("numReplicas", 0) >= 1
                    and conds.get("OptimizationReady") == "True"
                    and conds.get("MetricsAvailable") == "True"):
                return va
            return None

        va = _wait_for(optimized, 45, desc="desired allocation in VA status")
        st = va["status"]
        assert st["desiredOptimizedAlloc"]["accelerator"] == "A100"
        assert st["actuation"]["applied"] is True
        # collected load made it into currentAlloc (2-decimal strings)
        assert float(st["currentAlloc"]["load"]["arrivalRate"]) > 0
        assert st["currentAlloc"]["maxBatch"] == 256
        # ownerReference set from the deployment (controller.go:278-293)
        refs = va["metadata"].get("ownerReferences", [])
        assert refs and refs[0]["kind"] == "Deployment" and refs[0]["name"] == VA_NAME

        # -- controller /metrics exposes the inferno_* gauges ------------
        body = httpx.get(
            f"http://127.0.0.1:{world['metrics_port']}/metrics", timeout=5
        ).text
        assert "inferno_desired_replicas" in body
        assert f'variant_name="{VA_NAME}"' in body
        des_line = next(
            line for line in body.splitlines()
            if line.startswith("inferno_desired_replicas") and VA_NAME in line
        )
        assert float(des_line.rsplit(" ", 1)[-1]) >= 1.0

        # -- scale-in at idle (e2e_test.go:519): the [1m] rate window ----
        # empties after load stops; desired returns to the 1-replica floor
        def scaled_in():
            va = _va_status(kube)
            des = va["status"]["desiredOptimizedAlloc"]
            rate = float(va["status"]["currentAlloc"]["load"]["arrivalRate"] or 0)
            return va if (des["numReplicas"] <= 1 and rate < 0.5) else None

        _wait_for(scaled_in, 120, interval=2.0, desc="scale-in at idle")

    def test_config5_multi_va(self, world):
        """Second VA joins the fleet mid-flight (multi-VA scenario,
        e2e_test.go:698-1130 shape): both get optimized statuses."""
        kube = world["kube"]
        # second deployment+VA pointing at the same emulator metrics
        assert _apply(kube, {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": "vllme-deploy-b", "namespace": VA_NS},
            "spec": {"replicas": 1}, "status": {"replicas": 1},
        }).status_code == 201
        with open(os.path.join(REPO, "deploy", "examples",
                               "vllme-variantautoscaling.yaml")) as f:
            va_doc = next(d for d in yaml.safe_load_all(f)
                          if d and d["metadata"]["name"] == VA_NAME)
        va_doc["metadata"]["name"] = "vllme-deploy-b"
        va_doc["metadata"]["namespace"] = VA_NS
        assert _apply(kube, va_doc).status_code == 201

        def second_optimized():
            r = kube.get(
                f"/apis/llmd.ai/v1alpha1/namespaces/{VA_NS}"
                "/variantautoscalings/vllme-deploy-b"
            )
            if r.status_code != 200:
                return None
            st = r.json().get("status", {})
            conds = {c["type"]: c["status"] for c in st.get("conditions", [])}
            return r.json() if conds.get("OptimizationReady") == "True" else None

        _wait_for(second_optimized, 45, desc="second VA optimized")


@pytest.mark.e2e
class TestSecuredMetricsE2E:
    def test_controller_serves_https_metrics_with_token_auth(self, world, tmp_path):
        """A second controller instance with --metrics-cert-dir and a static
        token file: /metrics requires the bearer token over HTTPS
        (cmd/main.go:122-199 parity through the real process)."""
        crt, key = _mk_cert(tmp_path)
        cert_dir = tmp_path / "certs"
        cert_dir.mkdir()
        (cert_dir / "tls.crt").write_bytes(open(crt, "rb").read())
        (cert_dir / "tls.key").write_bytes(open(key, "rb").read())
        tok = tmp_path / "token"
        tok.write_text("s3cret-metrics\n")

        metrics_port = _free_port()
        proc = subprocess.Popen(
            [sys.executable, "-m", "inferno_amd.controller.main",
             "--metrics-port", str(metrics_port),
             "--metrics-cert-dir", str(cert_dir),
             "--metrics-auth-token-file", str(tok),
             "--no-leader-elect", "--backend", "cpu"],
            env={**os.environ,
                 "KUBE_API_URL": f"http://127.0.0.1:{world['api_port']}",
                 "PROMETHEUS_BASE_URL": world["prom_url"],
                 "PROMETHEUS_CA_CERT_PATH": world["prom_ca"]},
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, cwd=REPO,
        )
        try:
            base = f"https://127.0.0.1:{metrics_port}"
            import ssl

            ctx = ssl.create_default_context()
            ctx.check_hostname = False
            ctx.verify_mode = ssl.CERT_NONE
            deadline = time.time() + 30
            up = False
            while time.time() < deadline:
                if proc.poll() is not None:
                    pytest.fail("secured controller exited early")
                try:
                    r = httpx.get(f"{base}/healthz", verify=ctx, timeout=2)
                    if r.status_code == 200:
                        up = True
                        break
                except httpx.HTTPError:
                    time.sleep(0.3)
            assert up, "secured controller probes never came up"
            # unauthenticated scrape rejected; token accepted
            assert httpx.get(f"{base}/metrics", verify=ctx,
                             timeout=5).status_code == 401
            r = httpx.get(f"{base}/metrics", verify=ctx, timeout=5,
                          headers={"Authorization": "Bearer s3cret-metrics"})
            assert r.status_code == 200
            # plain HTTP against the TLS port fails
            with pytest.raises(httpx.HTTPError):
                httpx.get(f"http://127.0.0.1:{metrics_port}/healthz", timeout=2)
        finally:
            proc.send_signal(signal.SIGTERM)
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()


@pytest.mark.e2e
class TestTpVariantE2E:
    """The MI355X TP-variant path end to end: the llama70b example VA
    (perfmodel-derived MI355X/TP4/TP8 profiles, deploy/examples) reconciles
    through the real controller — TP degree as a first-class variant axis
    priced via the xGMI all-reduce model (BASELINE config 3 shape)."""

    def test_llama70b_tp_variants_optimized(self, world):
        kube = world["kube"]
        assert _apply(kube, {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": "llama70b-deploy", "namespace": VA_NS},
            "spec": {"replicas": 1}, "status": {"replicas": 1},
        }).status_code == 201
        with open(os.path.join(REPO, "deploy", "examples",
                               "vllme-variantautoscaling.yaml")) as f:
            va_doc = next(d for d in yaml.safe_load_all(f)
                          if d and d["metadata"]["name"] == "llama70b-deploy")
        va_doc["metadata"]["namespace"] = VA_NS
        # the emulator exports metrics for default/default only; point the
        # llama VA's modelID at it so the collector finds live load (the TP
        # profiles under test are unchanged)
        va_doc["spec"]["modelID"] = MODEL
        # keepAccelerator pins candidates to the labeled accelerator; label
        # the TP4 variant so the sweep prices accCount=4 over xGMI
        va_doc["metadata"]["labels"][
            "inference.optimization/acceleratorName"] = "MI355X-TP4"
        r = _apply(kube, va_doc)
        assert r.status_code == 201, r.text

        _drive_load(world["emu"], seconds=6.0, concurrency=4)

        def optimized():
            resp = kube.get(
                f"/apis/llmd.ai/v1alpha1/namespaces/{VA_NS}"
                "/variantautoscalings/llama70b-deploy")
            if resp.status_code != 200:
                return None
            st = resp.json().get("status", {})
            conds = {c["type"]: c["status"] for c in st.get("conditions", [])}
            des = st.get("desiredOptimizedAlloc", {})
            if conds.get("OptimizationReady") == "True" and des.get(
                    "numReplicas", 0) >= 1:
                return st
            return None

        st = _wait_for(optimized, 45, desc="llama70b TP-variant optimization")
        # keepAccelerator restricts the winner to the labeled TP variant
        assert st["desiredOptimizedAlloc"]["accelerator"] == "MI355X-TP4"


@pytest.mark.e2e
class TestHpaActuationLoop:
    """Close the actuation loop the reference's OpenShift tier asserts
    (sharegpt_scaleup_test.go:123-214): VA recommendation -> external-metric
    HPA (stand-in reading inferno_desired_replicas) -> Deployment replica
    convergence -> actuator's current_replicas gauge follows."""

    def test_recommendation_drives_deployment_replicas(self, world):
        from inferno_amd.testing.hpa import read_desired, reconcile_once

        kube = world["kube"]
        _drive_load(world["emu"], seconds=8.0, concurrency=8)

        metrics_url = f"http://127.0.0.1:{world['metrics_port']}/metrics"

        def desired_above_one():
            text = httpx.get(metrics_url, timeout=5).text
            d = read_desired(text).get((VA_NS, VA_NAME), 0)
            return d if d >= 2 else None

        desired = _wait_for(desired_above_one, 60, desc="scale-out recommendation")

        # one HPA pass applies the recommendation to the Deployment
        applied = reconcile_once(metrics_url, kube)
        assert (VA_NS, VA_NAME, desired) in applied
        dep = kube.get(
            f"/apis/apps/v1/namespaces/{VA_NS}/deployments/{VA_NAME}").json()
        assert dep["spec"]["replicas"] == desired

        # the controller's actuator reads REAL deployment replicas: the
        # current_replicas gauge converges to what HPA applied
        def current_follows():
            text = httpx.get(metrics_url, timeout=5).text
            for line in text.splitlines():
                if line.startswith("inferno_current_replicas") and VA_NAME in line:
                    if float(line.rsplit(" ", 1)[-1]) == float(desired):
                        return True
            return None

        _wait_for(current_follows, 30, interval=2.0,
                  desc="current_replicas gauge convergence")
