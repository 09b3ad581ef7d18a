"""Engine tests (CPU): snapshot construction + CPU sweep equals the direct
golden solve (Solver), unlimited winners agree."""
import numpy as np
import pytest

from inferno_amd.config import OptimizerSpec
from inferno_amd.core import System
from inferno_amd.engine import SweepEngine, build_cell_snapshot
from inferno_amd.solver import Solver
from tests.fixtures import make_spec


def build(**kw):
    spec = make_spec(**kw)
    system, opt = System.from_spec(spec)
    return system, opt


class TestSnapshot:
    def test_layout_and_segments(self):
        system, _ = build(n_servers=5, seed=41)
        snap = build_cell_snapshot(system)
        assert snap.n_cells == 5 * 3
        seg = snap.seg_start.numpy()
        assert seg[0] == 0 and seg[-1] == snap.n_cells
        assert np.all(np.diff(seg) == 3)
        # deterministic ordering: sorted server names x sorted acc names
        assert snap.server_names == sorted(system.servers)
        assert snap.cell_acc[:3] == sorted(["MI355X", "MI325X", "MI300X"])

    def test_keep_accelerator_restricts_cells(self):
        system, _ = build(n_servers=3, seed=42, keep_accelerator=True)
        snap = build_cell_snapshot(system)
        assert snap.n_cells == 3
        for seg, name in enumerate(snap.server_names):
            cur = system.servers[name].cur_allocation.accelerator
            assert snap.cell_acc[seg] == cur

    def test_missing_target_emits_no_cells(self):
        system, _ = build(n_servers=2, seed=43)
        srv = system.servers["srv-0:ns"]
        del system.service_classes[srv.service_class_name].targets[srv.model_name]
        snap = build_cell_snapshot(system)
        seg = snap.seg_start.numpy()
        assert seg[1] - seg[0] == 0  # srv-0 has no cells
        assert seg[2] - seg[1] == 3

    def test_flags_and_costs(self):
        system, _ = build(n_servers=1, seed=44)
        snap = build_cell_snapshot(system)
        srv = system.servers["srv-0:ns"]
        cur = srv.cur_allocation
        model = system.models[srv.model_name]
        for i, acc in enumerate(snap.cell_acc):
            flags = int(snap.arrays["flags"][i])
            assert bool(flags & 4)  # has cur
            assert bool(flags & 1) == (acc == cur.accelerator)
            want = system.accelerators[acc].cost * model.get_num_instances(acc)
            assert snap.arrays["acc_cost"][i].item() == pytest.approx(want, rel=1e-6)


class TestCpuEngineParity:
    def test_sweep_matches_direct_calculate(self):
        system_a, opt = build(n_servers=6, seed=45)
        system_b, _ = build(n_servers=6, seed=45)
        system_a.calculate()  # direct golden path
        SweepEngine(backend="cpu").sweep(system_b)
        for name in system_a.servers:
            a = system_a.servers[name].all_allocations
            b = system_b.servers[name].all_allocations
            assert set(a) == set(b)
            for acc in a:
                assert a[acc].num_replicas == b[acc].num_replicas
                assert a[acc].value == pytest.approx(b[acc].value, rel=1e-6)

    def test_solve_matches_reference_solver(self):
        system_a, opt = build(n_servers=8, seed=46)
        system_b, _ = build(n_servers=8, seed=46)
        system_a.calculate()
        Solver(opt).solve(system_a)
        SweepEngine(backend="cpu").solve(system_b, opt)
        for name in system_a.servers:
            a = system_a.servers[name].allocation
            b = system_b.servers[name].allocation
            assert (a is None) == (b is None)
            if a is not None:
                assert a.accelerator == b.accelerator
                assert a.num_replicas == b.num_replicas

    def test_greedy_mode_via_engine(self):
        cap = {"AMD-MI355X-288GB": 8, "AMD-MI325X-256GB": 8, "AMD-MI300X-192GB": 8}
        system, opt = build(n_servers=6, seed=47, unlimited=False, capacity=cap)
        stats = SweepEngine(backend="cpu").solve(system, opt)
        assert stats.n_cells == 18
        used = {t: 0 for t in cap}
        for srv in system.servers.values():
            if srv.allocation is None:
                continue
            acc = system.accelerators[srv.allocation.accelerator]
            model = system.models[srv.model_name]
            used[acc.type] += (
                srv.allocation.num_replicas
                * model.get_num_instances(acc.name)
                * acc.multiplicity
            )
        for t in cap:
            assert used[t] <= cap[t]

    def test_zero_load_scale_to_zero_winner(self):
        from inferno_amd.config import ServerLoadSpec

        system, opt = build(n_servers=2, seed=48, min_num_replicas=0)
        system.servers["srv-0:ns"].load = ServerLoadSpec(0.0, 0, 0)
        SweepEngine(backend="cpu").solve(system, opt)
        alloc = system.servers["srv-0:ns"].allocation
        assert alloc is not None
        # empty allocation has strongly negative value (-0.9*cur cost), wins
        assert alloc.accelerator == ""
        assert alloc.num_replicas == 0


class TestHugeBuckets:
    """choose_buckets dispatches N > 8192 cells to the global-memory spill
    bucket (uncapped sweep)."""

    def test_bucket_partition_xl_and_gmem(self):
        import numpy as np

        from inferno_amd.ops.sweep import MAX_N, XL_MAX_N, choose_buckets

        batch_n = np.array([8, 600, 3000, 9000, 20000, 40000], dtype=np.int32)
        buckets = choose_buckets(batch_n)
        gmem_buckets = [b for b in buckets if b[4]]
        lds_buckets = [b for b in buckets if not b[4]]
        # 40000 > XL_MAX_N -> global-memory spill
        assert len(gmem_buckets) == 1
        nt, ids, bmax, count, gmem = gmem_buckets[0]
        assert count == 1 and bmax == 40000 and set(ids) == {5}
        # 9000/20000 -> XL LDS tier (160KB dynamic-LDS opt-in)
        xl = [b for b in lds_buckets if b[2] > MAX_N]
        assert len(xl) == 1
        nt, ids, bmax, count, gmem = xl[0]
        assert count == 2 and bmax == 20000 and set(ids) == {3, 4}
        assert bmax <= XL_MAX_N
        for _, _, bmax, _, _ in lds_buckets:
            assert bmax <= XL_MAX_N

    def test_xl_override_forces_gmem(self, monkeypatch):
        import numpy as np

        import inferno_amd.ops.sweep as sweep

        monkeypatch.setattr(sweep, "XL_MAX_N", sweep.MAX_N)
        buckets = sweep.choose_buckets(np.array([9000], dtype=np.int32))
        assert len(buckets) == 1 and buckets[0][4] is True

    def test_absurd_n_fails_loudly(self):
        import numpy as np
        import pytest as _pytest

        from inferno_amd.ops.sweep import HUGE_MAX_N, HipKernelError, choose_buckets

        with _pytest.raises(HipKernelError):
            choose_buckets(np.array([HUGE_MAX_N + 1], dtype=np.int64))

    def test_fastpath_batch_uncapped(self):
        """FastSweep's vectorized batch sizing must agree with the scalar
        compute_batch_size at huge N (no clamp on either path)."""
        from inferno_amd.core.system import System
        from inferno_amd.engine.fastpath import FastSweep
        from inferno_amd.engine.snapshot import compute_batch_size
        from tests.fixtures import make_spec

        from inferno_amd.config import ServerLoadSpec

        system, _ = System.from_spec(make_spec(n_servers=2, seed=77))
        for srv in system.servers.values():
            srv.load = ServerLoadSpec(arrivalRate=600.0, avgInTokens=64, avgOutTokens=3)
        for model in system.models.values():
            for perf in model.perf_data.values():
                perf.maxBatchSize = 256
                perf.atTokens = 1024
        fs = FastSweep(system, backend="cpu")
        arrs = fs._refresh_dynamic()
        want = 256 * 1024 // 3
        assert want > 8192
        assert (arrs["batch_n"] == want).all()
        srv = next(iter(system.servers.values()))
        perf = system.models[srv.model_name].get_perf_data("MI355X")
        assert compute_batch_size(srv, perf, 3) == want
