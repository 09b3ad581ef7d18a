"""Multi-GPU solver sharding over RCCL/xGMI (gloo on CPU for tests).

Servers are partitioned round-robin across ranks; each rank sweeps and
argmin-solves only its shard (HIP kernels on its own MI355X), then the
tiny per-server winner records are all-gathered and the per-accelerator-type
aggregates accumulated. Payloads are KBs — latency-bound on xGMI — so a
single fixed-size all_gather (one collective) is used rather than chatty
object collectives (SURVEY.md section 5 "distributed communication backend").

The reference has no distributed path (single Go process); this implements
the sharded equivalent of SolveUnlimited + AllocateByType + GenerateSolution
with identical results to a single-process solve.

Hot path: FastSweep (engine/fastpath.py) — static cell structure cached per
fleet topology, vectorized dynamic refresh, winners-only materialization.
Greedy (limited) mode falls back to the full SweepEngine candidate lists.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from ..config import AllocationData, OptimizerSpec, SaturationPolicy
from ..core.system import AllocationByType, System
from ..engine import SweepEngine
from ..engine.fastpath import FastSweep, WinnerRecord

# winner record encoding: [server_global_idx, acc_code, num_replicas, cost,
#                          batch, itl, ttft, valid]
_REC_W = 8
_ACC_EMPTY = -2.0  # zero-load empty allocation (accelerator "")
_ACC_NONE = -1.0  # no feasible allocation


@dataclass
class ShardResult:
    solution: dict[str, AllocationData]
    allocation_by_type: dict[str, AllocationByType]
    local_stats: object


def shard_servers(all_names: list[str], rank: int, world: int) -> list[str]:
    return all_names[rank::world]


@dataclass
class _ShardStats:
    n_cells: int = 0
    n_servers: int = 0


class ShardedSolver:
    """Data-parallel sweep across ranks with all-gathered winners."""

    def __init__(self, engine: SweepEngine, group=None, fast: bool = True):
        self.engine = engine
        self.group = group
        self.fast = fast
        self._fast_sweep: Optional[FastSweep] = None
        self._fast_key = None

    def invalidate(self) -> None:
        """Drop the cached cell structure (fleet topology changed)."""
        self._fast_sweep = None
        self._fast_key = None

    def _dist_info(self):
        import torch.distributed as dist

        initialized = dist.is_available() and dist.is_initialized()
        rank = dist.get_rank(self.group) if initialized else 0
        world = dist.get_world_size(self.group) if initialized else 1
        return initialized, rank, world

    # ------------------------------------------------------------------
    def solve(self, system: System, spec: OptimizerSpec) -> ShardResult:
        import torch
        import torch.distributed as dist

        initialized, rank, world = self._dist_info()
        all_names = sorted(system.servers)
        acc_names = sorted(system.accelerators)
        local_names = shard_servers(all_names, rank, world)

        if spec.unlimited and self.fast:
            rec, stats = self._solve_fast(system, local_names)
        else:
            rec, stats = self._solve_slow(system, local_names, spec, acc_names)

        # encode local winners into a fixed-size record tensor
        max_shard = (len(all_names) + world - 1) // world
        rec_t = torch.full((max_shard, _REC_W), -3.0, dtype=torch.float32)
        name_to_global = {n: i for i, n in enumerate(all_names)}
        import numpy as np

        n_local = len(local_names)
        if n_local:
            gidx = np.asarray([name_to_global[n] for n in local_names], dtype=np.float32)
            body = np.stack(
                [
                    gidx,
                    rec.acc_idx.astype(np.float32),
                    rec.num_replicas.astype(np.float32),
                    rec.cost,
                    rec.batch.astype(np.float32),
                    rec.itl,
                    rec.ttft,
                    np.ones(n_local, dtype=np.float32),
                ],
                axis=1,
            )
            rec_t[:n_local] = torch.from_numpy(body)

        if initialized and world > 1:
            backend = dist.get_backend(self.group)
            comm_dev = "cuda" if backend == "nccl" else "cpu"
            rec_d = rec_t.to(comm_dev)
            gathered = [torch.empty_like(rec_d) for _ in range(world)]
            dist.all_gather(gathered, rec_d, group=self.group)
            all_rec = torch.cat(gathered, dim=0).cpu().numpy()
        else:
            all_rec = rec_t.numpy()

        # reconstruct the global solution (identical on every rank)
        solution: dict[str, AllocationData] = {}
        by_type: dict[str, AllocationByType] = {}
        for row in all_rec:
            if row[7] != 1.0:
                continue
            name = all_names[int(row[0])]
            code = row[1]
            if code == _ACC_NONE:
                continue
            acc_name = "" if code == _ACC_EMPTY else acc_names[int(code)]
            server = system.servers[name]
            data = AllocationData(
                accelerator=acc_name,
                numReplicas=int(row[2]),
                maxBatch=int(row[4]),
                cost=float(row[3]),
                itlAverage=float(row[5]),
                ttftAverage=float(row[6]),
                load=server.load,
            )
            solution[name] = data
            if acc_name:
                acc = system.accelerators.get(acc_name)
                model = system.models.get(server.model_name)
                if acc is not None and model is not None:
                    t = acc.type
                    agg = by_type.setdefault(
                        t, AllocationByType(name=t, limit=system.capacity.get(t, 0))
                    )
                    agg.count += (
                        data.numReplicas * model.get_num_instances(acc.name) * acc.multiplicity
                    )
                    agg.cost += data.cost

        system.allocation_by_type = by_type
        return ShardResult(solution=solution, allocation_by_type=by_type, local_stats=stats)

    # ------------------------------------------------------------------
    def _solve_fast(self, system: System, local_names: list[str]):
        key = (id(system), tuple(local_names))
        if self._fast_key != key:
            self._fast_sweep = FastSweep(
                system,
                local_names,
                backend=self.engine.backend,
                device=self.engine.device,
            )
            self._fast_key = key
        rec = self._fast_sweep.reconcile()
        stats = _ShardStats(n_cells=self._fast_sweep.n_cells, n_servers=len(local_names))
        return rec, stats

    def _solve_slow(self, system: System, local_names, spec: OptimizerSpec, acc_names):
        stats = self.engine.sweep(system, server_names=local_names)
        if spec.unlimited:
            for name in local_names:
                server = system.servers[name]
                server.remove_allocation()
                best = None
                for an in sorted(server.all_allocations):
                    alloc = server.all_allocations[an]
                    if best is None or alloc.value < best.value:
                        best = alloc
                if best is not None:
                    server.set_allocation(best)
        else:
            # greedy limited mode needs the full candidate lists; capacity is
            # global, so greedy over a shard is only exact for world==1 —
            # multi-rank greedy runs rank-local greedy on the shard with the
            # full capacity map (documented approximation).
            from ..solver.greedy import solve_greedy

            solve_greedy(
                system,
                delayed_best_effort=spec.delayedBestEffort,
                saturation_policy=SaturationPolicy.parse(spec.saturationPolicy),
            )

        acc_index = {n: i for i, n in enumerate(acc_names)}
        n_local = len(local_names)
        from ..engine.fastpath import _empty_winner

        rec = _empty_winner(n_local)
        for j, name in enumerate(local_names):
            alloc = system.servers[name].allocation
            if alloc is None:
                continue
            rec.acc_idx[j] = -2 if alloc.accelerator == "" else acc_index[alloc.accelerator]
            rec.num_replicas[j] = alloc.num_replicas
            rec.batch[j] = alloc.batch_size
            rec.cost[j] = alloc.cost
            rec.itl[j] = alloc.itl
            rec.ttft[j] = alloc.ttft
        return rec, stats
