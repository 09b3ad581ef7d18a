#!/usr/bin/env python3
"""Long-run hardware soak of the ENTIRE framework: the real controller with
the HIP solver + apiserver stand-in + TLS Prometheus stand-in + emulator +
HPA stand-in, under burst/idle load cycles on an MI355X.

  python scripts/gpu_stack_soak.py [--minutes 10] [--backend gpu]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import httpx

from inferno_amd.testing.stack import VA_NAME, VA_NS, LocalStack


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=10.0)
    p.add_argument("--backend", default="gpu", choices=["auto", "gpu", "cpu"])
    args = p.parse_args()

    stack = LocalStack(interval="2s", backend=args.backend, with_hpa=True,
                       quiet=True)
    try:
        stack.up()
        print(f"stack up ({args.backend} backend)", flush=True)
        t_end = time.time() + args.minutes * 60
        cycle = 0
        peak = 0
        while time.time() < t_end:
            cycle += 1
            stack.drive_load(12.0, concurrency=6)
            time.sleep(18)
            dead = [i for i, pr in enumerate(stack.procs)
                    if pr.poll() is not None]
            assert not dead, f"processes died: {dead}"
            dep = stack.kube.get(
                f"/apis/apps/v1/namespaces/{VA_NS}/deployments/{VA_NAME}"
            ).json()
            st = stack.va_status()
            m = httpx.get(f"http://127.0.0.1:{stack.metrics_port}/metrics",
                          timeout=5)
            assert m.status_code == 200
            replicas = dep["spec"]["replicas"]
            desired = st["desiredOptimizedAlloc"]["numReplicas"]
            peak = max(peak, replicas)
            if cycle % 4 == 0 or cycle <= 2:
                print(f"cycle {cycle}: replicas={replicas} desired={desired}",
                      flush=True)
        print(f"STACK SOAK OK: {cycle} cycles, peak replicas {peak}, "
              f"all processes alive", flush=True)
        return 0
    finally:
        stack.down()


if __name__ == "__main__":
    raise SystemExit(main())
