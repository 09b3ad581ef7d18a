"""Model-analyzer / optimizer-engine adapter parity tests
(ref internal/modelanalyzer + internal/optimizer envtest coverage)."""
import pytest

from inferno_amd.api import v1alpha1 as api
from inferno_amd.config import OptimizerSpec
from inferno_amd.controller.modelanalyzer import (
    ModelAnalyzer,
    VariantAutoscalingsEngine,
)
from inferno_amd.core import System
from inferno_amd.engine import SweepEngine
from inferno_amd.solver import Manager, Optimizer
from tests.fixtures import make_spec


def build(**kw):
    spec = make_spec(**kw)
    return System.from_spec(spec)


def va_for(server_name: str) -> api.VariantAutoscaling:
    name, ns = server_name.split(":")
    return api.VariantAutoscaling(name=name, namespace=ns)


class TestModelAnalyzer:
    def test_analyze_model_returns_candidates(self):
        system, _ = build(n_servers=3, seed=71)
        ma = ModelAnalyzer(system)
        resp = ma.analyze_model(va_for("srv-0:ns"))
        assert set(resp.allocations) == {"MI355X", "MI325X", "MI300X"}
        for acc, entry in resp.allocations.items():
            assert entry.allocation.num_replicas >= 1
            assert entry.reason == "markovian analysis"

    def test_unknown_server_empty_response(self):
        system, _ = build(n_servers=1, seed=72)
        resp = ModelAnalyzer(system).analyze_model(va_for("nope:ns"))
        assert resp.allocations == {}


class TestOptimizerEngine:
    def test_optimize_maps_by_va_name(self):
        system, opt = build(n_servers=4, seed=73)
        system.calculate()
        engine = VariantAutoscalingsEngine(Manager(system, Optimizer(opt)), system)
        vas = [va_for(n) for n in sorted(system.servers)]
        out = engine.optimize(vas)
        assert set(out) == {va.name for va in vas}
        for va in vas:
            alloc = out[va.name]
            server = system.servers[f"{va.name}:{va.namespace}"]
            assert alloc.accelerator == server.allocation.accelerator
            assert alloc.numReplicas == server.allocation.num_replicas
            assert alloc.lastRunTime  # timestamped

    def test_no_solution_raises(self):
        system, opt = build(n_servers=1, seed=74)
        # no calculate() -> no candidates -> empty solution
        engine = VariantAutoscalingsEngine(Manager(system, Optimizer(opt)), system)
        with pytest.raises(RuntimeError):
            engine.optimize([va_for("srv-0:ns")])


class TestLeaderElection:
    def test_fail_open_without_lease_api(self):
        from inferno_amd.controller.k8s import InMemoryKube
        from inferno_amd.controller.leader import LeaderElector

        elector = LeaderElector(InMemoryKube(), "test-lease", "ns", "me")
        assert elector.try_acquire() is True  # in-memory fake: no election
