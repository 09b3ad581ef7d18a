"""Multi-GPU solver sharding over RCCL/xGMI (gloo on CPU for tests).

Servers are partitioned round-robin across ranks; each rank sweeps and
argmin-solves only its shard (HIP kernels on its own MI355X), then the
tiny per-server winner records are all-gathered and the per-accelerator-type
aggregates all-reduced. Payloads are KBs — latency-bound on xGMI — so a
single fixed-size all_gather (one collective) is used rather than chatty
object collectives (SURVEY.md section 5 "distributed communication backend").

The reference has no distributed path (single Go process); this implements
the sharded equivalent of SolveUnlimited + AllocateByType + GenerateSolution
with identical results to a single-process solve.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from ..config import AllocationData, OptimizerSpec
from ..core.system import AllocationByType, System
from ..engine import SweepEngine

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


class ShardedSolver:
    """Data-parallel sweep across ranks with all-gathered winners."""

    def __init__(self, engine: SweepEngine, group=None):
        self.engine = engine
        self.group = group

    def solve(self, system: System, spec: OptimizerSpec) -> ShardResult:
        import torch
        import torch.distributed as dist

        initialized = dist.is_available() and dist.is_initialized()
        rank = dist.get_rank(self.group) if initialized else 0
        world = dist.get_world_size(self.group) if initialized else 1

        all_names = sorted(system.servers)
        acc_names = sorted(system.accelerators)
        acc_index = {n: i for i, n in enumerate(acc_names)}
        local_names = shard_servers(all_names, rank, world)

        stats = self.engine.sweep(system, server_names=local_names)
        # local unlimited argmin over the shard
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

        # encode local winners
        max_shard = (len(all_names) + world - 1) // world
        device = "cuda" if (self.engine.backend == "gpu") else "cpu"
        rec = torch.full((max_shard, _REC_W), -3.0, dtype=torch.float32)
        name_to_global = {n: i for i, n in enumerate(all_names)}
        for j, name in enumerate(local_names):
            server = system.servers[name]
            alloc = server.allocation
            rec[j, 0] = float(name_to_global[name])
            if alloc is None:
                rec[j, 1] = _ACC_NONE
                rec[j, 7] = 1.0
                continue
            rec[j, 1] = float(acc_index[alloc.accelerator]) if alloc.accelerator else _ACC_EMPTY
            rec[j, 2] = float(alloc.num_replicas)
            rec[j, 3] = alloc.cost
            rec[j, 4] = float(alloc.batch_size)
            rec[j, 5] = alloc.itl
            rec[j, 6] = alloc.ttft
            rec[j, 7] = 1.0

        if initialized and world > 1:
            backend = dist.get_backend(self.group)
            comm_dev = device if backend == "nccl" else "cpu"
            rec_d = rec.to(comm_dev)
            gathered = [torch.empty_like(rec_d) for _ in range(world)]
            dist.all_gather(gathered, rec_d, group=self.group)
            all_rec = torch.cat(gathered, dim=0).cpu()
        else:
            all_rec = rec

        # reconstruct global solution + apply remote winners to local view
        solution: dict[str, AllocationData] = {}
        for row in all_rec:
            if row[7].item() != 1.0:
                continue
            gidx = int(row[0].item())
            name = all_names[gidx]
            code = row[1].item()
            if code == _ACC_NONE:
                continue
            acc = "" if code == _ACC_EMPTY else acc_names[int(code)]
            server = system.servers[name]
            data = AllocationData(
                accelerator=acc,
                numReplicas=int(row[2].item()),
                maxBatch=int(row[4].item()),
                cost=float(row[3].item()),
                itlAverage=float(row[5].item()),
                ttftAverage=float(row[6].item()),
                load=server.load,
            )
            solution[name] = data

        # per-type aggregation (AllocateByType semantics) over the GLOBAL
        # solution — identical on every rank, so no extra collective needed;
        # counts need model numInstances which every rank has (full registry).
        by_type: dict[str, AllocationByType] = {}
        for name, data in solution.items():
            if not data.accelerator:
                continue
            server = system.servers[name]
            acc = system.accelerators.get(data.accelerator)
            model = system.models.get(server.model_name)
            if acc is None or model is None:
                continue
            t = acc.type
            agg = by_type.setdefault(
                t, AllocationByType(name=t, limit=system.capacity.get(t, 0))
            )
            agg.count += data.numReplicas * model.get_num_instances(acc.name) * acc.multiplicity
            agg.cost += data.cost
        system.allocation_by_type = by_type
        return ShardResult(solution=solution, allocation_by_type=by_type, local_stats=stats)
