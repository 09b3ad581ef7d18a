"""Native C++ greedy (ops/native/greedy.cpp) vs the Python golden
(solver/greedy.py) — differential over randomized limited fleets, all
saturation policies. Host-only code: runs without a GPU (the .so's greedy
entry point is pure CPU)."""
import numpy as np
import pytest

from inferno_amd.config import SaturationPolicy, ServerLoadSpec
from inferno_amd.core.system import System
from inferno_amd.engine import SweepEngine
from inferno_amd.engine.fastpath import FastSweep
from inferno_amd.solver.greedy import solve_greedy
from tests.fixtures import make_spec

pytest.importorskip("inferno_amd.ops.sweep")


def _require_lib():
    from inferno_amd.ops.sweep import load_library

    try:
        load_library(allow_build=True)
    except Exception as e:  # pragma: no cover
        pytest.skip(f"native library unavailable: {e}")


def cells_from_cpu_sweep(system, local_names):
    """Produce the GPU-sweep-shaped per-cell output arrays from the CPU
    golden sweep, so the native greedy can be differentially tested without
    a GPU."""
    SweepEngine(backend="cpu").sweep(system, server_names=local_names)
    fs = FastSweep(system, local_names, backend="cpu")
    n = fs.n_cells
    cells = {
        "feasible": np.zeros(n, np.uint8),
        "zero_empty": np.zeros(n, np.uint8),
        "num_replicas": np.zeros(n, np.int32),
        "batch": np.zeros(n, np.int32),
        "cost": np.zeros(n, np.float32),
        "value": np.zeros(n, np.float32),
        "itl": np.zeros(n, np.float32),
        "ttft": np.zeros(n, np.float32),
        "rho": np.zeros(n, np.float32),
        "max_rate": np.zeros(n, np.float32),
        "cell_server": fs.cell_server,
        "cell_acc_idx": fs.cell_acc_idx,
        "seg_start": fs.seg_start,
    }
    for k in range(n):
        srv = system.servers[local_names[fs.cell_server[k]]]
        acc_name = fs.acc_names[fs.cell_acc_idx[k]]
        alloc = srv.all_allocations.get(acc_name)
        if alloc is None:
            continue
        cells["feasible"][k] = 1
        cells["zero_empty"][k] = 1 if alloc.accelerator == "" else 0
        cells["num_replicas"][k] = alloc.num_replicas
        cells["batch"][k] = alloc.batch_size
        cells["cost"][k] = alloc.cost
        cells["value"][k] = alloc.value
        cells["itl"][k] = alloc.itl
        cells["ttft"][k] = alloc.ttft
        cells["rho"][k] = alloc.rho
        cells["max_rate"][k] = alloc.max_arrv_rate_per_replica
    return fs, cells


def run_native(system, spec, cells, fs, local_names):
    from inferno_amd.parallel.dist import ShardedSolver

    solver = ShardedSolver(SweepEngine(backend="cpu"))
    solver._fast_sweep = fs
    solver._fast_key = (id(system), tuple(local_names))
    acc_names = sorted(system.accelerators)
    rec = solver._solve_limited_native(system, local_names, cells, spec, acc_names)
    assert rec is not None, "native greedy path unavailable"
    return rec


POLICIES = ["None", "PriorityExhaustive", "PriorityRoundRobin", "RoundRobin"]


class TestNativeGreedyDifferential:
    @pytest.mark.parametrize("policy", POLICIES)
    @pytest.mark.parametrize("seed", [0, 1, 2])
    def test_matches_python_golden(self, policy, seed):
        _require_lib()
        rng = np.random.default_rng(seed)
        cap = {
            "AMD-MI300X-192GB": int(rng.integers(2, 30)),
            "AMD-MI325X-256GB": int(rng.integers(2, 30)),
            "AMD-MI355X-288GB": int(rng.integers(2, 30)),
        }
        kw = dict(n_servers=24, seed=700 + seed, unlimited=False,
                  capacity=dict(cap), saturation_policy=policy)
        a, opt = System.from_spec(make_spec(**kw))
        b, _ = System.from_spec(make_spec(**kw))
        local_names = sorted(a.servers)

        # golden: CPU sweep + Python greedy
        SweepEngine(backend="cpu").sweep(b)
        solve_greedy(b, delayed_best_effort=opt.delayedBestEffort,
                     saturation_policy=SaturationPolicy.parse(opt.saturationPolicy))

        # native: same candidate set through the C++ solver
        fs, cells = cells_from_cpu_sweep(a, local_names)
        run_native(a, opt, cells, fs, local_names)

        for name in local_names:
            ga, gb = a.servers[name].allocation, b.servers[name].allocation
            assert (ga is None) == (gb is None), f"{name}: allocated mismatch"
            if ga is None:
                continue
            assert ga.accelerator == gb.accelerator, name
            assert ga.num_replicas == gb.num_replicas, name
            assert ga.cost == pytest.approx(gb.cost, rel=1e-6), name
            assert ga.value == pytest.approx(gb.value, rel=1e-6), name

    def test_delayed_best_effort(self):
        _require_lib()
        cap = {"AMD-MI300X-192GB": 6, "AMD-MI325X-256GB": 6, "AMD-MI355X-288GB": 6}
        kw = dict(n_servers=16, seed=711, unlimited=False, capacity=dict(cap),
                  saturation_policy="RoundRobin", delayed_best_effort=True)
        a, opt = System.from_spec(make_spec(**kw))
        b, _ = System.from_spec(make_spec(**kw))
        local_names = sorted(a.servers)
        SweepEngine(backend="cpu").sweep(b)
        solve_greedy(b, delayed_best_effort=True,
                     saturation_policy=SaturationPolicy.ROUND_ROBIN)
        fs, cells = cells_from_cpu_sweep(a, local_names)
        run_native(a, opt, cells, fs, local_names)
        for name in local_names:
            ga, gb = a.servers[name].allocation, b.servers[name].allocation
            assert (ga is None) == (gb is None), name
            if ga is not None:
                assert (ga.accelerator, ga.num_replicas) == (
                    gb.accelerator, gb.num_replicas), name

    def test_zero_load_servers_dropped(self):
        _require_lib()
        cap = {"AMD-MI300X-192GB": 10, "AMD-MI325X-256GB": 10,
               "AMD-MI355X-288GB": 10}
        kw = dict(n_servers=8, seed=712, unlimited=False, capacity=dict(cap),
                  min_num_replicas=0)
        a, opt = System.from_spec(make_spec(**kw))
        b, _ = System.from_spec(make_spec(**kw))
        for s in (a, b):
            for i, n in enumerate(sorted(s.servers)):
                if i % 2 == 0:
                    s.servers[n].load = ServerLoadSpec(0.0, 0, 0)
        local_names = sorted(a.servers)
        SweepEngine(backend="cpu").sweep(b)
        solve_greedy(b)
        fs, cells = cells_from_cpu_sweep(a, local_names)
        run_native(a, opt, cells, fs, local_names)
        for name in local_names:
            ga, gb = a.servers[name].allocation, b.servers[name].allocation
            assert (ga is None) == (gb is None), name
            if ga is not None:
                assert (ga.accelerator, ga.num_replicas) == (
                    gb.accelerator, gb.num_replicas), name

    def test_capacity_never_exceeded(self):
        _require_lib()
        cap = {"AMD-MI300X-192GB": 5, "AMD-MI325X-256GB": 5, "AMD-MI355X-288GB": 5}
        kw = dict(n_servers=32, seed=713, unlimited=False, capacity=dict(cap),
                  saturation_policy="PriorityRoundRobin")
        a, opt = System.from_spec(make_spec(**kw))
        local_names = sorted(a.servers)
        fs, cells = cells_from_cpu_sweep(a, local_names)
        run_native(a, opt, cells, fs, local_names)
        used = {t: 0 for t in cap}
        for n in local_names:
            alloc = a.servers[n].allocation
            if alloc is None or alloc.accelerator == "":
                continue
            acc = a.accelerators[alloc.accelerator]
            model = a.models[a.servers[n].model_name]
            used[acc.type] += (alloc.num_replicas
                               * model.get_num_instances(acc.name)
                               * acc.multiplicity)
        for t in cap:
            assert used[t] <= cap[t], t


class TestNativeGreedyFuzz:
    """Wider randomized differential: many fleets/capacity regimes in one
    test (cheap: CPU sweep reused per seed)."""

    @pytest.mark.parametrize("seed", range(3, 13))
    def test_random_regimes(self, seed):
        _require_lib()
        rng = np.random.default_rng(1000 + seed)
        policy = POLICIES[seed % 4]
        cap = {
            "AMD-MI300X-192GB": int(rng.integers(0, 50)),
            "AMD-MI325X-256GB": int(rng.integers(0, 50)),
            "AMD-MI355X-288GB": int(rng.integers(0, 50)),
        }
        kw = dict(n_servers=int(rng.integers(4, 20)), seed=2000 + seed,
                  unlimited=False, capacity=dict(cap), saturation_policy=policy,
                  delayed_best_effort=bool(seed % 2),
                  priorities=(1, 5, 10) if seed % 3 == 0 else (1, 10))
        a, opt = System.from_spec(make_spec(**kw))
        b, _ = System.from_spec(make_spec(**kw))
        local_names = sorted(a.servers)
        SweepEngine(backend="cpu").sweep(b)
        solve_greedy(b, delayed_best_effort=opt.delayedBestEffort,
                     saturation_policy=SaturationPolicy.parse(opt.saturationPolicy))
        fs, cells = cells_from_cpu_sweep(a, local_names)
        run_native(a, opt, cells, fs, local_names)
        for name in local_names:
            ga, gb = a.servers[name].allocation, b.servers[name].allocation
            assert (ga is None) == (gb is None), f"{name} ({policy})"
            if ga is not None:
                assert (ga.accelerator, ga.num_replicas) == (
                    gb.accelerator, gb.num_replicas), f"{name} ({policy})"
