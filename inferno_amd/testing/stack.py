"""One-command local bring-up of the full autoscaler stack.

The executable counterpart of the reference's `deploy/install.sh` Kind
bring-up for an image with no cluster binaries: starts the kube-apiserver
stand-in, the vLLM emulator, the TLS Prometheus stand-in and the REAL
controller process; applies the CRD, the three ConfigMaps, a variant
Deployment and the sample VariantAutoscaling; waits for the controller to
become ready and (optionally) for the first optimized status.

Run:  python -m inferno_amd.testing.stack [--duration 0] [--smoke]
      (duration 0 = run until Ctrl-C; --smoke = wait for an optimized VA,
       print a summary and exit 0/1)
"""
from __future__ import annotations

import argparse
import os
import signal
import socket
import subprocess
import sys
import tempfile
import threading
import time

import httpx
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
NS_SYS = "workload-variant-autoscaler-system"
VA_NS = "llm-d-sim"
VA_NAME = "vllme-deploy"
MODEL = "default/default"
CRD = os.path.join(REPO, "deploy", "crd", "llmd.ai_variantautoscalings.yaml")


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _spawn_announced(cmd, env=None, match="listening on"):
    e = dict(os.environ)
    if env:
        e.update(env)
    proc = subprocess.Popen(cmd, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                            env=e, text=True, cwd=REPO)
    port = None
    deadline = time.time() + 30
    while time.time() < deadline:
        line = proc.stdout.readline()
        if match in line:
            port = int(line.strip().rsplit(" ", 1)[-1])
            break
        if proc.poll() is not None:
            break
    if port is None:
        err = proc.stderr.read() if proc.poll() is not None else "(no announce)"
        proc.kill()
        raise RuntimeError(f"failed to start {cmd[:3]}...: {err[-1500:]}")
    threading.Thread(target=lambda: [None for _ in proc.stdout], daemon=True).start()
    threading.Thread(target=lambda: [None for _ in proc.stderr], daemon=True).start()
    return proc, port


def _mk_cert(dirpath: str):
    crt = os.path.join(dirpath, "tls.crt")
    key = os.path.join(dirpath, "tls.key")
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", key, "-out", crt, "-days", "7", "-subj", "/CN=promstub",
         "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    return crt, key


class LocalStack:
    """Owns the four processes + seeded objects."""

    def __init__(self, interval: str = "5s", backend: str = "auto",
                 quiet: bool = False, with_hpa: bool = False):
        self.interval = interval
        self.backend = backend
        self.quiet = quiet
        self.with_hpa = with_hpa
        self.procs: list[subprocess.Popen] = []
        self.kube: httpx.Client | None = None
        self.api_port = self.emu_port = self.prom_port = self.metrics_port = 0
        self._tmp = tempfile.mkdtemp(prefix="wva-stack-")

    def _log(self, msg: str):
        if not self.quiet:
            print(f"[stack] {msg}", flush=True)

    def apply(self, doc: dict):
        kind = doc["kind"]
        base = {
            "ConfigMap": ("api/v1", "configmaps"),
            "Deployment": ("apis/apps/v1", "deployments"),
            "VariantAutoscaling": ("apis/llmd.ai/v1alpha1", "variantautoscalings"),
        }[kind]
        ns = doc.get("metadata", {}).get("namespace", "default")
        r = self.kube.post(f"/{base[0]}/namespaces/{ns}/{base[1]}", json=doc)
        if r.status_code not in (200, 201, 409):
            raise RuntimeError(f"apply {kind}/{doc['metadata'].get('name')}: "
                               f"{r.status_code} {r.text[:300]}")
        return r

    def up(self):
        # 1. apiserver
        p, self.api_port = _spawn_announced(
            [sys.executable, "-m", "inferno_amd.testing.kubeapi", "--port", "0",
             "--crd", CRD], match="kubeapi listening on")
        self.procs.append(p)
        self.kube = httpx.Client(base_url=f"http://127.0.0.1:{self.api_port}",
                                 timeout=10.0)
        self._log(f"kube-apiserver stand-in on :{self.api_port} (CRD applied)")

        # 2. emulator
        self.emu_port = _free_port()
        p = subprocess.Popen(
            [sys.executable, "-m", "inferno_amd.emulator.server"],
            env={**os.environ, "PORT": str(self.emu_port), "MODEL_NAME": MODEL,
                 "NAMESPACE": VA_NS, "DECODE_TIME": "2", "PREFILL_TIME": "2",
                 "MAX_BATCH_SIZE": "256"},
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, cwd=REPO)
        self.procs.append(p)
        emu = f"http://127.0.0.1:{self.emu_port}"
        for _ in range(100):
            try:
                if httpx.get(f"{emu}/healthz", timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.2)
        else:
            raise RuntimeError("emulator did not become ready")
        self._log(f"vLLM emulator on :{self.emu_port} (model {MODEL})")

        # 3. TLS prometheus stand-in
        self.ca_cert, key = _mk_cert(self._tmp)
        p, self.prom_port = _spawn_announced(
            [sys.executable, "-m", "inferno_amd.testing.promstub", "--port", "0",
             "--target", emu, "--interval", "1.0",
             "--tls-cert", self.ca_cert, "--tls-key", key],
            match="promstub listening on")
        self.procs.append(p)
        self._log(f"prometheus stand-in on :{self.prom_port} (TLS)")

        # 4. cluster objects
        for f in ("configmap-accelerator-unitcost.yaml", "configmap-serviceclass.yaml"):
            for doc in yaml.safe_load_all(open(os.path.join(REPO, "deploy", f))):
                if doc:
                    self.apply(doc)
        self.apply({
            "apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {
                "name": "workload-variant-autoscaler-variantautoscaling-config",
                "namespace": NS_SYS},
            "data": {"GLOBAL_OPT_INTERVAL": self.interval},
        })
        self.apply({
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": VA_NAME, "namespace": VA_NS},
            "spec": {"replicas": 1}, "status": {"replicas": 1},
        })
        with open(os.path.join(REPO, "deploy", "examples",
                               "vllme-variantautoscaling.yaml")) as f:
            va_doc = next(d for d in yaml.safe_load_all(f)
                          if d and d["metadata"]["name"] == VA_NAME)
        va_doc["metadata"]["namespace"] = VA_NS
        self.apply(va_doc)
        self._log("applied ConfigMaps, variant Deployment and sample VA")

        # 5. controller
        self.metrics_port = _free_port()
        p = subprocess.Popen(
            [sys.executable, "-m", "inferno_amd.controller.main",
             "--metrics-port", str(self.metrics_port), "--backend", self.backend],
            env={**os.environ,
                 "KUBE_API_URL": f"http://127.0.0.1:{self.api_port}",
                 "PROMETHEUS_BASE_URL": f"https://127.0.0.1:{self.prom_port}",
                 "PROMETHEUS_CA_CERT_PATH": self.ca_cert},
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, cwd=REPO)
        self.procs.append(p)
        for _ in range(150):
            try:
                if httpx.get(f"http://127.0.0.1:{self.metrics_port}/readyz",
                             timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            if p.poll() is not None:
                raise RuntimeError("controller exited during startup")
            time.sleep(0.2)
        else:
            raise RuntimeError("controller never became ready")
        self._log(f"controller ready; metrics/probes on :{self.metrics_port}")

        # 6. optional HPA stand-in: closes the external-metric actuation loop
        # (inferno_desired_replicas -> Deployment replicas)
        if self.with_hpa:
            p = subprocess.Popen(
                [sys.executable, "-m", "inferno_amd.testing.hpa",
                 "--metrics-url",
                 f"http://127.0.0.1:{self.metrics_port}/metrics",
                 "--kube-url", f"http://127.0.0.1:{self.api_port}",
                 "--interval", "2"],
                stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, cwd=REPO)
            self.procs.append(p)
            self._log("HPA stand-in running (external-metric actuation loop)")

    def drive_load(self, seconds: float, concurrency: int = 6) -> int:
        emu = f"http://127.0.0.1:{self.emu_port}"
        stop = time.time() + seconds
        done = [0]
        lock = threading.Lock()

        def worker():
            with httpx.Client(timeout=30.0) as c:
                while time.time() < stop:
                    try:
                        c.post(f"{emu}/v1/chat/completions", json={
                            "model": MODEL,
                            "messages": [{"role": "user", "content": "hi " * 64}],
                            "max_tokens": 32})
                        with lock:
                            done[0] += 1
                    except httpx.HTTPError:
                        pass

        ts = [threading.Thread(target=worker) for _ in range(concurrency)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        return done[0]

    def va_status(self) -> dict:
        r = self.kube.get(
            f"/apis/llmd.ai/v1alpha1/namespaces/{VA_NS}/variantautoscalings/{VA_NAME}")
        r.raise_for_status()
        return r.json().get("status", {})

    def down(self):
        if self.kube is not None:
            self.kube.close()
        for p in self.procs:
            try:
                p.send_signal(signal.SIGTERM)
            except OSError:
                pass
        for p in self.procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        self.procs.clear()


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--duration", type=float, default=0.0,
                   help="seconds to run (0 = until interrupted)")
    p.add_argument("--smoke", action="store_true",
                   help="drive load, wait for an optimized VA status, exit")
    p.add_argument("--interval", default="5s")
    p.add_argument("--backend", default="auto", choices=["auto", "gpu", "cpu"])
    p.add_argument("--with-hpa", action="store_true",
                   help="run the HPA stand-in so recommendations actually "
                        "scale the Deployment")
    args = p.parse_args()

    stack = LocalStack(interval=args.interval, backend=args.backend,
                       with_hpa=args.with_hpa)
    try:
        stack.up()
        if args.smoke:
            n = stack.drive_load(10.0)
            print(f"[stack] drove {n} requests through the emulator", flush=True)
            deadline = time.time() + 60
            while time.time() < deadline:
                st = stack.va_status()
                conds = {c["type"]: c["status"] for c in st.get("conditions", [])}
                if (st.get("desiredOptimizedAlloc", {}).get("numReplicas", 0) >= 1
                        and conds.get("OptimizationReady") == "True"):
                    des = st["desiredOptimizedAlloc"]
                    print(f"[stack] SMOKE PASS: desired {des['numReplicas']} "
                          f"replicas on {des['accelerator']}", flush=True)
                    return 0
                time.sleep(2)
            print("[stack] SMOKE FAIL: VA never optimized", flush=True)
            return 1
        print(f"[stack] running; controller metrics: "
              f"http://127.0.0.1:{stack.metrics_port}/metrics  "
              f"(Ctrl-C to stop)", flush=True)
        deadline = time.time() + args.duration if args.duration else None
        try:
            while deadline is None or time.time() < deadline:
                time.sleep(1)
        except KeyboardInterrupt:
            pass
        return 0
    finally:
        stack.down()


if __name__ == "__main__":
    raise SystemExit(main())
