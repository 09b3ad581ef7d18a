#!/usr/bin/env python3
"""Controller-level differential soak: run the FULL reconcile loop (ConfigMaps
-> collector -> adapters -> solver -> status apply -> metrics) for many ticks
against an emulated Prometheus whose load follows a bursty trace, with a CPU
and a GPU reconciler fed identical worlds, and compare every desired
allocation they write.

  python scripts/controller_soak.py --vas 64 --ticks 200
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from prometheus_client import CollectorRegistry

from inferno_amd.api import v1alpha1 as api
from inferno_amd.controller import collector
from inferno_amd.controller.collector import MockPromAPI, Sample
from inferno_amd.controller.k8s import Deployment, InMemoryKube
from inferno_amd.controller.metrics import MetricsEmitter
from inferno_amd.controller.reconciler import Reconciler

NS = "workload-variant-autoscaler-system"
ACC_CM = {
    "MI300X": json.dumps({"device": "AMD-MI300X-192GB", "cost": "65.00"}),
    "MI325X": json.dumps({"device": "AMD-MI325X-256GB", "cost": "78.00"}),
    "MI355X": json.dumps({"device": "AMD-MI355X-288GB", "cost": "95.00"}),
}


def make_world(n_vas, rng):
    kube = InMemoryKube()
    kube.add_configmap(NS, "accelerator-unit-costs", ACC_CM)
    rows = []
    for i in range(n_vas):
        rows.append(f"  - model: m{i}\n    slo-tpot: {rng.integers(20, 200)}\n"
                    f"    slo-ttft: {rng.integers(500, 3000)}")
    kube.add_configmap(NS, "service-classes-config",
                       {"premium.yaml": "name: Premium\npriority: 1\ndata:\n"
                        + "\n".join(rows)})
    kube.add_configmap(NS, "workload-variant-autoscaler-variantautoscaling-config",
                       {"GLOBAL_OPT_INTERVAL": "60s"})
    accs = list(ACC_CM)
    for i in range(n_vas):
        acc = accs[i % 3]
        va = api.VariantAutoscaling(
            name=f"va-{i}", namespace="default",
            labels={api.ACCELERATOR_LABEL: acc},
            spec=api.VariantAutoscalingSpec(
                modelID=f"m{i}",
                sloClassRef=api.ConfigMapKeyRef("service-classes-config", "premium.yaml"),
                modelProfile=api.ModelProfile(accelerators=[
                    api.AcceleratorProfile(
                        acc=acc, accCount=1,
                        perfParms=api.PerfParms(
                            decodeParms={"alpha": f"{rng.uniform(5, 60):.3f}",
                                         "beta": f"{rng.uniform(0.05, 2):.4f}"},
                            prefillParms={"gamma": f"{rng.uniform(1, 30):.3f}",
                                          "delta": f"{rng.uniform(0.001, 0.2):.5f}"},
                        ),
                        maxBatchSize=int(rng.integers(4, 256)),
                    )
                ]),
            ),
        )
        kube.add_va(va)
        kube.add_deployment(Deployment(name=va.name, namespace="default",
                                       replicas=1, status_replicas=1, uid=f"uid-{i}"))
    return kube


def prom_for(n_vas, rates, rng_tok):
    now = time.time()
    results = {}
    for i in range(n_vas):
        m = f"m{i}"
        results[collector.arrival_query(m, "default")] = [Sample(rates[i], now)]
        results[collector.ttft_query(m, "default")] = [Sample(0.05, now)]
        results[collector.itl_query(m, "default")] = [Sample(0.01, now)]
        results[collector.avg_prompt_tokens_query(m, "default")] = [
            Sample(int(rng_tok[i, 0]), now)]
        results[collector.avg_decode_tokens_query(m, "default")] = [
            Sample(int(rng_tok[i, 1]), now)]
    return MockPromAPI(results=results)


class LatencyProm:
    """MockPromAPI wrapper simulating network latency per query (the
    full-reconcile-loop bench: 5 PromQL + availability probe per VA,
    ref controller.go:86-201)."""

    def __init__(self, inner, latency_ms: float):
        self.inner = inner
        self.latency_s = latency_ms / 1000.0

    def query(self, promql):
        if self.latency_s > 0:
            time.sleep(self.latency_s)
        return self.inner.query(promql)


def bench_full_loop(args, rng):
    """--bench mode: time the FULL reconcile loop —
    prepare -> collect (mocked-latency PromQL) -> solve -> status write —
    at fleet scale on the requested backend; report p50/p95."""
    kube = make_world(args.vas, np.random.default_rng(args.seed))
    rec = Reconciler(kube, None, MetricsEmitter(registry=CollectorRegistry()),
                     backend=args.backend, scale_to_zero=False)
    tok = np.random.default_rng(args.seed + 1).integers(8, 2048, size=(args.vas, 2))
    lat = []
    for t in range(args.ticks):
        rates = rng.gamma(1.5, 2.0, args.vas) * rng.choice(
            [0.0, 0.2, 1.0, 5.0], args.vas, p=[0.1, 0.3, 0.4, 0.2])
        rec.prom = LatencyProm(prom_for(args.vas, rates, tok), args.prom_latency_ms)
        t0 = time.perf_counter()
        res = rec.reconcile()
        lat.append((time.perf_counter() - t0) * 1000)
        if res.errors[:1] and t == 0:
            print("tick errors:", res.errors[:3])
    lat.sort()
    p50 = lat[len(lat) // 2]
    p95 = lat[max(int(len(lat) * 0.95) - 1, 0)]
    print(f"full-loop bench: vas={args.vas} ticks={args.ticks} "
          f"backend={rec.engine.backend} prom_latency_ms={args.prom_latency_ms} "
          f"collect_workers={rec.collect_workers} "
          f"p50={p50:.2f}ms p95={p95:.2f}ms")
    return 0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--vas", type=int, default=64)
    p.add_argument("--ticks", type=int, default=200)
    p.add_argument("--seed", type=int, default=20260913)
    p.add_argument("--bench", action="store_true",
                   help="time the full reconcile loop only (no differential)")
    p.add_argument("--backend", choices=["auto", "gpu", "cpu"], default="auto")
    p.add_argument("--prom-latency-ms", type=float, default=0.0,
                   help="simulated per-PromQL-query latency")
    args = p.parse_args()
    rng = np.random.default_rng(args.seed)
    if args.bench:
        return bench_full_loop(args, rng)

    worlds = {}
    recs = {}
    for backend in ("cpu", "gpu"):
        kube = make_world(args.vas, np.random.default_rng(args.seed))
        recs[backend] = Reconciler(
            kube, None, MetricsEmitter(registry=CollectorRegistry()),
            backend=backend, scale_to_zero=False)
        worlds[backend] = kube

    tok = np.random.default_rng(args.seed + 1).integers(8, 2048, size=(args.vas, 2))
    lat = []
    mismatches = 0
    decisions = 0
    for t in range(args.ticks):
        # bursty rates, occasional zero-load
        rates = rng.gamma(1.5, 2.0, args.vas) * rng.choice(
            [0.0, 0.2, 1.0, 5.0], args.vas, p=[0.1, 0.3, 0.4, 0.2])
        prom = prom_for(args.vas, rates, tok)
        out = {}
        for backend in ("cpu", "gpu"):
            recs[backend].prom = prom
            t0 = time.perf_counter()
            recs[backend].reconcile()
            dt = time.perf_counter() - t0
            if backend == "gpu":
                lat.append(dt * 1000)
            out[backend] = {
                name: (va.status.desiredOptimizedAlloc.accelerator,
                       va.status.desiredOptimizedAlloc.numReplicas)
                for (ns, name), va in worlds[backend].vas.items()
            }
        for name in out["cpu"]:
            decisions += 1
            a, b = out["cpu"][name], out["gpu"][name]
            if a[0] != b[0] or abs(a[1] - b[1]) > 1:
                mismatches += 1
                if mismatches <= 5:
                    print(f"DECISION MISMATCH tick={t} {name}: cpu={a} gpu={b}")
    lat.sort()
    print(f"controller soak: vas={args.vas} ticks={args.ticks} decisions={decisions} "
          f"mismatches={mismatches} gpu_tick_p50={lat[len(lat)//2]*1:.2f}ms "
          f"p95={lat[int(len(lat)*0.95)-1]:.2f}ms")
    print("CONTROLLER SOAK PASS" if mismatches == 0 else "CONTROLLER SOAK FAIL")
    return 0 if mismatches == 0 else 1


if __name__ == "__main__":
    raise SystemExit(main())
