"""Multi-process sharded-solver tests (gloo backend, world_size=2, CPU).

Verifies the distributed path is correct by construction: a 2-rank sharded
solve produces exactly the same global solution as a single-process solve.
"""
import json
import multiprocessing as mp
import os

import pytest


def _worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.parallel import ShardedSolver
    from tests.fixtures import make_spec

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    try:
        spec = make_spec(n_servers=9, seed=77)
        system, opt = System.from_spec(spec)
        solver = ShardedSolver(SweepEngine(backend="cpu"))
        result = solver.solve(system, opt)
        payload = {
            name: (d.accelerator, d.numReplicas, round(d.cost, 4))
            for name, d in result.solution.items()
        }
        by_type = {
            t: (a.count, round(a.cost, 3)) for t, a in result.allocation_by_type.items()
        }
        q.put((rank, json.dumps(payload, sort_keys=True), json.dumps(by_type, sort_keys=True)))
    finally:
        dist.destroy_process_group()


class TestShardedSolver:
    def test_two_rank_solution_matches_single(self):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        port = 29815
        procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(2):
            rank, payload, by_type = q.get(timeout=120)
            results[rank] = (payload, by_type)
        for p in procs:
            p.join(timeout=30)
            assert p.exitcode == 0

        # both ranks agree on the global solution
        assert results[0] == results[1]

        # and match the single-process solve
        from inferno_amd.core.system import System
        from inferno_amd.engine import SweepEngine
        from inferno_amd.parallel import ShardedSolver
        from tests.fixtures import make_spec

        spec = make_spec(n_servers=9, seed=77)
        system, opt = System.from_spec(spec)
        result = ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
        single = {
            name: (d.accelerator, d.numReplicas, round(d.cost, 4))
            for name, d in result.solution.items()
        }
        assert json.loads(results[0][0]) == {k: list(v) for k, v in single.items()}

    def test_shard_partitioning(self):
        from inferno_amd.parallel import shard_servers

        names = [f"s{i}" for i in range(10)]
        shards = [shard_servers(names, r, 4) for r in range(4)]
        assert sorted(sum(shards, [])) == sorted(names)
        assert all(len(s) in (2, 3) for s in shards)


def _greedy_worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.parallel import ShardedSolver
    from tests.fixtures import make_spec

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    try:
        cap = {"AMD-MI355X-288GB": 8, "AMD-MI325X-256GB": 8, "AMD-MI300X-192GB": 8}
        spec = make_spec(n_servers=9, seed=78, unlimited=False, capacity=cap)
        system, opt = System.from_spec(spec)
        result = ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
        payload = {
            name: (d.accelerator, d.numReplicas) for name, d in result.solution.items()
        }
        q.put((rank, json.dumps(payload, sort_keys=True)))
    finally:
        dist.destroy_process_group()


class TestShardedGreedy:
    def test_two_rank_greedy_matches_single(self):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        port = 29817
        procs = [ctx.Process(target=_greedy_worker, args=(r, 2, port, q)) for r in range(2)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(2):
            rank, payload = q.get(timeout=120)
            results[rank] = payload
        for p in procs:
            p.join(timeout=30)
            assert p.exitcode == 0
        assert results[0] == results[1]  # both ranks agree globally

        from inferno_amd.core.system import System
        from inferno_amd.engine import SweepEngine
        from inferno_amd.parallel import ShardedSolver
        from tests.fixtures import make_spec

        cap = {"AMD-MI355X-288GB": 8, "AMD-MI325X-256GB": 8, "AMD-MI300X-192GB": 8}
        spec = make_spec(n_servers=9, seed=78, unlimited=False, capacity=cap)
        system, opt = System.from_spec(spec)
        result = ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
        single = {
            name: (d.accelerator, d.numReplicas) for name, d in result.solution.items()
        }
        assert json.loads(results[0]) == {k: list(v) for k, v in single.items()}


class TestShardedSolverWorld3:
    """Non-power-of-2 world size: 9 servers over 3 ranks (uneven shards are
    exercised by 4 servers on rank 0 vs 2 on rank 2 with 10 servers)."""

    def test_three_rank_solution_matches_single(self):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        port = 29819
        procs = [ctx.Process(target=_worker3, args=(r, 3, port, q)) for r in range(3)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(3):
            rank, payload = q.get(timeout=120)
            results[rank] = payload
        for p in procs:
            p.join(timeout=30)
            assert p.exitcode == 0
        assert results[0] == results[1] == results[2]

        from inferno_amd.core.system import System
        from inferno_amd.engine import SweepEngine
        from inferno_amd.parallel import ShardedSolver
        from tests.fixtures import make_spec

        spec = make_spec(n_servers=10, seed=91)
        system, opt = System.from_spec(spec)
        result = ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
        single = {
            name: (d.accelerator, d.numReplicas, round(d.cost, 4))
            for name, d in result.solution.items()
        }
        assert json.loads(results[0]) == {k: list(v) for k, v in single.items()}


def _worker3(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.parallel import ShardedSolver
    from tests.fixtures import make_spec

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    try:
        spec = make_spec(n_servers=10, seed=91)
        system, opt = System.from_spec(spec)
        result = ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
        payload = {
            name: (d.accelerator, d.numReplicas, round(d.cost, 4))
            for name, d in result.solution.items()
        }
        q.put((rank, json.dumps(payload, sort_keys=True)))
    finally:
        dist.destroy_process_group()
