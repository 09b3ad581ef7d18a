"""Failure-detection & resilience: GPU solver failure -> CPU fallback with
SolverDegraded reason; engine reentrancy under concurrent solves (the
reference relies on serialized reconciles because of its global singletons —
this build is reentrant by design, SURVEY.md section 5 race-detection plan)."""
import json
import threading
import time

import pytest
from prometheus_client import CollectorRegistry

from inferno_amd.api import v1alpha1 as api
from inferno_amd.controller import collector
from inferno_amd.controller.collector import MockPromAPI, Sample
from inferno_amd.controller.k8s import Deployment, InMemoryKube
from inferno_amd.controller.metrics import MetricsEmitter
from inferno_amd.controller.reconciler import Reconciler
from tests.test_controller import ACCELERATOR_CM, NS, SERVICE_CLASS_CM, make_va


def build_world():
    kube = InMemoryKube()
    kube.add_configmap(NS, "accelerator-unit-costs", ACCELERATOR_CM)
    kube.add_configmap(NS, "service-classes-config", SERVICE_CLASS_CM)
    kube.add_configmap(NS, "workload-variant-autoscaler-variantautoscaling-config",
                       {"GLOBAL_OPT_INTERVAL": "60s"})
    va = make_va()
    kube.add_va(va)
    kube.add_deployment(Deployment(name=va.name, namespace=va.namespace, replicas=1,
                                   status_replicas=1, uid="u1"))
    now = time.time()
    model = "default/default"
    prom = MockPromAPI(
        results={
            collector.arrival_query(model, "default"): [Sample(2.0, now)],
            collector.ttft_query(model, "default"): [Sample(0.05, now)],
            collector.itl_query(model, "default"): [Sample(0.01, now)],
            collector.avg_prompt_tokens_query(model, "default"): [Sample(128, now)],
            collector.avg_decode_tokens_query(model, "default"): [Sample(64, now)],
        }
    )
    em = MetricsEmitter(registry=CollectorRegistry())
    return kube, prom, em


class _ExplodingSolver:
    def __init__(self):
        self.calls = 0

    def solve(self, system, spec):
        self.calls += 1
        raise RuntimeError("simulated HIP device loss")


class TestSolverDegraded:
    def test_gpu_failure_falls_back_to_cpu_with_condition(self):
        kube, prom, em = build_world()
        rec = Reconciler(kube, prom, em, backend="cpu", scale_to_zero=False)
        # pretend we were on GPU and the solver dies mid-tick
        rec.engine.backend = "gpu"
        rec.solver = _ExplodingSolver()
        result = rec.reconcile()
        assert result.processed == 1
        assert result.degraded is True
        assert result.solver_backend == "cpu"
        va = kube.vas[("default", "vllme-deploy")]
        cond = api.get_condition(va, api.TYPE_OPTIMIZATION_READY)
        assert cond.status == "True"
        assert cond.reason == api.REASON_SOLVER_DEGRADED
        # the reconciler replaced the dead solver with a working CPU one
        assert rec.engine.backend == "cpu"

    def test_cpu_failure_marks_optimization_failed(self):
        kube, prom, em = build_world()
        rec = Reconciler(kube, prom, em, backend="cpu", scale_to_zero=False)
        rec.solver = _ExplodingSolver()
        result = rec.reconcile()
        assert result.processed == 0
        va = kube.vas[("default", "vllme-deploy")]
        cond = api.get_condition(va, api.TYPE_OPTIMIZATION_READY)
        assert cond.status == "False"
        assert cond.reason == api.REASON_OPTIMIZATION_FAILED


class TestReentrancy:
    def test_concurrent_solves_are_isolated(self):
        """Two systems solved on two threads concurrently produce the same
        results as solved sequentially (no shared mutable globals)."""
        from inferno_amd.core import System
        from inferno_amd.engine import SweepEngine
        from tests.fixtures import make_spec

        def solve(seed):
            system, opt = System.from_spec(make_spec(n_servers=6, seed=seed))
            SweepEngine(backend="cpu").solve(system, opt)
            return {
                n: (s.allocation.accelerator, s.allocation.num_replicas)
                for n, s in system.servers.items()
                if s.allocation is not None
            }

        sequential = [solve(901), solve(902)]
        results = [None, None]
        errors = []

        def worker(i, seed):
            try:
                results[i] = solve(seed)
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        threads = [
            threading.Thread(target=worker, args=(0, 901)),
            threading.Thread(target=worker, args=(1, 902)),
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=120)
        assert not errors
        assert results == sequential
