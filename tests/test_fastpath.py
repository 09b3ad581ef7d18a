"""FastSweep (cached snapshot + winners-only) correctness vs the full path."""
import numpy as np
import pytest

from inferno_amd.config import ServerLoadSpec
from inferno_amd.core import System, allocation_from_data
from inferno_amd.engine import FastSweep, SweepEngine, build_cell_snapshot
from inferno_amd.parallel import ShardedSolver
from tests.fixtures import make_spec


def build(**kw):
    spec = make_spec(**kw)
    return System.from_spec(spec)


class TestSnapshotEquivalence:
    def test_cell_arrays_match_full_builder(self):
        system, _ = build(n_servers=6, seed=61)
        fast = FastSweep(system, backend="cpu")
        snap = build_cell_snapshot(system)
        arrs = fast.cell_arrays()
        assert fast.n_cells == snap.n_cells
        for k, v in snap.arrays.items():
            got = arrs[k].numpy()
            want = v.numpy()
            np.testing.assert_array_equal(got, want, err_msg=k)

    def test_refresh_tracks_mutations(self):
        system, _ = build(n_servers=4, seed=62)
        fast = FastSweep(system, backend="cpu")
        # mutate loads and current allocations
        system.servers["srv-0:ns"].load = ServerLoadSpec(999.0, 77, 55)
        from inferno_amd.config import AllocationData

        system.servers["srv-1:ns"].cur_allocation = allocation_from_data(
            AllocationData(accelerator="MI355X", numReplicas=9, cost=123.0)
        )
        snap = build_cell_snapshot(system)
        arrs = fast.cell_arrays()
        for k, v in snap.arrays.items():
            np.testing.assert_array_equal(arrs[k].numpy(), v.numpy(), err_msg=k)

    def test_zero_load_and_keep_accelerator(self):
        system, _ = build(n_servers=4, seed=63, keep_accelerator=True, min_num_replicas=0)
        system.servers["srv-2:ns"].load = ServerLoadSpec(0.0, 0, 0)
        fast = FastSweep(system, backend="cpu")
        snap = build_cell_snapshot(system)
        arrs = fast.cell_arrays()
        assert fast.n_cells == snap.n_cells == 4
        for k, v in snap.arrays.items():
            np.testing.assert_array_equal(arrs[k].numpy(), v.numpy(), err_msg=k)


class TestFastCpuWinners:
    def test_matches_sweep_engine_solve(self):
        system_a, opt = build(n_servers=8, seed=64)
        system_b, _ = build(n_servers=8, seed=64)
        SweepEngine(backend="cpu").solve(system_a, opt)
        fast = FastSweep(system_b, backend="cpu")
        rec = fast.reconcile()
        for seg, name in enumerate(fast.server_names):
            alloc = system_a.servers[name].allocation
            if alloc is None:
                assert rec.acc_idx[seg] == -1
                continue
            if alloc.accelerator == "":
                assert rec.acc_idx[seg] == -2
            else:
                assert fast.acc_names[rec.acc_idx[seg]] == alloc.accelerator
            assert rec.num_replicas[seg] == alloc.num_replicas
            assert rec.cost[seg] == pytest.approx(alloc.cost, rel=1e-6)


class TestShardedFastVsSlow:
    def test_fast_equals_slow_single_rank(self):
        system_a, opt = build(n_servers=10, seed=65)
        system_b, _ = build(n_servers=10, seed=65)
        fast = ShardedSolver(SweepEngine(backend="cpu"), fast=True).solve(system_a, opt)
        slow = ShardedSolver(SweepEngine(backend="cpu"), fast=False).solve(system_b, opt)
        assert set(fast.solution) == set(slow.solution)
        for name in fast.solution:
            f, s = fast.solution[name], slow.solution[name]
            assert f.accelerator == s.accelerator
            assert f.numReplicas == s.numReplicas
            assert f.cost == pytest.approx(s.cost, rel=1e-6)
        assert set(fast.allocation_by_type) == set(slow.allocation_by_type)
        for t in fast.allocation_by_type:
            assert fast.allocation_by_type[t].count == slow.allocation_by_type[t].count

    def test_cache_reuse_and_invalidate(self):
        system, opt = build(n_servers=4, seed=66)
        solver = ShardedSolver(SweepEngine(backend="cpu"), fast=True)
        solver.solve(system, opt)
        first = solver._fast_sweep
        solver.solve(system, opt)
        assert solver._fast_sweep is first  # cached
        solver.invalidate()
        solver.solve(system, opt)
        assert solver._fast_sweep is not first
