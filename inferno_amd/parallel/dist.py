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
The global solution is carried as numpy arrays (``GlobalWinners``); the
per-server ``AllocationData`` dict is materialized lazily only when a
consumer (the controller's status writer) asks for it. Greedy (limited)
mode falls back to the full SweepEngine candidate lists.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from functools import cached_property
from typing import Optional

import numpy as np

from ..config import AllocationData, OptimizerSpec, SaturationPolicy
from ..core.system import AllocationByType, System
from ..engine import SweepEngine
from ..engine.fastpath import FastSweep, _empty_winner

# winner record encoding: [server_global_idx, acc_code, num_replicas, cost,
#                          batch, itl, ttft, valid]
_REC_W = 8
_ACC_EMPTY = -2.0  # zero-load empty allocation (accelerator "")
_ACC_NONE = -1.0  # no feasible allocation


def shard_servers(all_names: list[str], rank: int, world: int) -> list[str]:
    return all_names[rank::world]


@dataclass
class _ShardStats:
    n_cells: int = 0
    n_servers: int = 0


@dataclass
class GlobalWinners:
    """Global solution as flat arrays (index = position in sorted names)."""

    names: list[str]
    acc_names: list[str]
    acc_idx: np.ndarray  # int32: accelerator index, -1 none, -2 empty
    num_replicas: np.ndarray  # int32
    batch: np.ndarray  # int32
    cost: np.ndarray  # float32
    itl: np.ndarray
    ttft: np.ndarray
    valid: np.ndarray  # bool: a rank reported this server


@dataclass
class ShardResult:
    winners: GlobalWinners
    allocation_by_type: dict[str, AllocationByType]
    local_stats: object
    _system: Optional[System] = field(default=None, repr=False)

    @cached_property
    def solution(self) -> dict[str, AllocationData]:
        """Per-server AllocationData map (GenerateSolution wire shape);
        materialized lazily from the winner arrays."""
        w = self.winners
        out: dict[str, AllocationData] = {}
        for i, name in enumerate(w.names):
            if not w.valid[i]:
                continue
            code = int(w.acc_idx[i])
            if code == -1:
                continue
            acc = "" if code == -2 else w.acc_names[code]
            load = None
            if self._system is not None:
                srv = self._system.servers.get(name)
                load = srv.load if srv is not None else None
            data = AllocationData(
                accelerator=acc,
                numReplicas=int(w.num_replicas[i]),
                maxBatch=int(w.batch[i]),
                cost=float(w.cost[i]),
                itlAverage=float(w.itl[i]),
                ttftAverage=float(w.ttft[i]),
            )
            if load is not None:
                data.load = load
            out[name] = data
        return out


class ShardedSolver:
    """Data-parallel sweep across ranks with all-gathered winners."""

    def __init__(self, engine: SweepEngine, group=None, fast: bool = True):
        self.engine = engine
        self.group = group
        self.fast = fast
        self._fast_sweep: Optional[FastSweep] = None
        self._fast_key = None
        self._agg_key = None
        self._agg = None  # cached aggregation structures

    def invalidate(self) -> None:
        """Drop cached structures (fleet topology changed)."""
        self._fast_sweep = None
        self._fast_key = None
        self._agg_key = None
        self._agg = None

    @property
    def fast_sweep(self) -> Optional[FastSweep]:
        return self._fast_sweep

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
        n_local = len(local_names)
        body = np.full((max_shard, _REC_W), -3.0, dtype=np.float32)
        if n_local:
            gidx = np.arange(rank, rank + world * n_local, world, dtype=np.float32)
            body[:n_local, 0] = gidx[:n_local]
            body[:n_local, 1] = rec.acc_idx
            body[:n_local, 2] = rec.num_replicas
            body[:n_local, 3] = rec.cost
            body[:n_local, 4] = rec.batch
            body[:n_local, 5] = rec.itl
            body[:n_local, 6] = rec.ttft
            body[:n_local, 7] = 1.0

        if initialized and world > 1:
            backend = dist.get_backend(self.group)
            comm_dev = "cuda" if backend == "nccl" else "cpu"
            rec_d = torch.from_numpy(body).to(comm_dev)
            gathered = [torch.empty_like(rec_d) for _ in range(world)]
            dist.all_gather(gathered, rec_d, group=self.group)
            all_rec = torch.cat(gathered, dim=0).cpu().numpy()
        else:
            all_rec = body

        # scatter rows back into global order (vectorized)
        n = len(all_names)
        winners = GlobalWinners(
            names=all_names,
            acc_names=acc_names,
            acc_idx=np.full(n, -1, dtype=np.int32),
            num_replicas=np.zeros(n, dtype=np.int32),
            batch=np.zeros(n, dtype=np.int32),
            cost=np.zeros(n, dtype=np.float32),
            itl=np.zeros(n, dtype=np.float32),
            ttft=np.zeros(n, dtype=np.float32),
            valid=np.zeros(n, dtype=bool),
        )
        ok = all_rec[:, 7] == 1.0
        rows = all_rec[ok]
        g = rows[:, 0].astype(np.int64)
        winners.acc_idx[g] = rows[:, 1].astype(np.int32)
        winners.num_replicas[g] = rows[:, 2].astype(np.int32)
        winners.batch[g] = rows[:, 4].astype(np.int32)
        winners.cost[g] = rows[:, 3]
        winners.itl[g] = rows[:, 5]
        winners.ttft[g] = rows[:, 6]
        winners.valid[g] = True

        by_type = self._aggregate_by_type(system, all_names, acc_names, winners)
        system.allocation_by_type = by_type
        return ShardResult(
            winners=winners,
            allocation_by_type=by_type,
            local_stats=stats,
            _system=system,
        )

    # ------------------------------------------------------------------
    def _aggregate_by_type(self, system, all_names, acc_names, winners):
        """Vectorized AllocateByType (ref system.go:271-300)."""
        key = (id(system), tuple(acc_names), len(all_names))
        if self._agg_key != key:
            n_acc = len(acc_names)
            inst = np.zeros((len(all_names), n_acc), dtype=np.int32)
            for i, name in enumerate(all_names):
                model = system.models.get(system.servers[name].model_name)
                if model is None:
                    continue
                for j, an in enumerate(acc_names):
                    inst[i, j] = model.get_num_instances(an)
            mult = np.array(
                [system.accelerators[a].multiplicity for a in acc_names], dtype=np.int32
            )
            types = [system.accelerators[a].type for a in acc_names]
            uniq_types = sorted(set(types))
            type_idx = np.array([uniq_types.index(t) for t in types], dtype=np.int32)
            self._agg = (inst, mult, types, uniq_types, type_idx)
            self._agg_key = key
        inst, mult, types, uniq_types, type_idx = self._agg

        sel = winners.valid & (winners.acc_idx >= 0)
        idx = np.nonzero(sel)[0]
        by_type: dict[str, AllocationByType] = {}
        if len(idx):
            codes = winners.acc_idx[idx]
            counts = (
                winners.num_replicas[idx] * inst[idx, codes] * mult[codes]
            ).astype(np.int64)
            tcount = np.zeros(len(uniq_types), dtype=np.int64)
            tcost = np.zeros(len(uniq_types), dtype=np.float64)
            np.add.at(tcount, type_idx[codes], counts)
            np.add.at(tcost, type_idx[codes], winners.cost[idx].astype(np.float64))
            for k, t in enumerate(uniq_types):
                if tcount[k] == 0 and tcost[k] == 0.0:
                    continue
                by_type[t] = AllocationByType(
                    name=t,
                    count=int(tcount[k]),
                    limit=system.capacity.get(t, 0),
                    cost=float(tcost[k]),
                )
        return by_type

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

    def _populate_from_cells(self, system: System, local_names, cells) -> None:
        """Fill server.all_allocations from per-cell GPU output arrays (only
        feasible cells are materialized as objects)."""
        from ..core import Allocation

        acc_names = sorted(system.accelerators)
        for name in local_names:
            system.servers[name].all_allocations = {}
        idx = np.nonzero(cells["feasible"])[0]
        if not len(idx):
            return
        # convert columns to Python scalars once (tolist) instead of paying
        # the numpy-scalar box/convert cost per field per cell
        srv_i = cells["cell_server"][idx].tolist()
        acc_i = cells["cell_acc_idx"][idx].tolist()
        zero = cells["zero_empty"][idx].tolist()
        reps = cells["num_replicas"][idx].tolist()
        batch = cells["batch"][idx].tolist()
        cost = cells["cost"][idx].tolist()
        value = cells["value"][idx].tolist()
        itl = cells["itl"][idx].tolist()
        ttft = cells["ttft"][idx].tolist()
        rho = cells["rho"][idx].tolist()
        max_rate = cells["max_rate"][idx].tolist()
        servers = system.servers
        for k in range(len(idx)):
            acc_key = acc_names[acc_i[k]]
            servers[local_names[srv_i[k]]].all_allocations[acc_key] = Allocation(
                accelerator="" if zero[k] else acc_key,
                num_replicas=reps[k],
                batch_size=batch[k],
                cost=cost[k],
                value=value[k],
                itl=itl[k],
                ttft=ttft[k],
                rho=rho[k],
                max_arrv_rate_per_replica=max_rate[k],
            )

    def _solve_slow(self, system: System, local_names, spec: OptimizerSpec, acc_names):
        # greedy limited mode on GPU: candidates from the cached FastSweep
        # (K4 plan: GPU sweep produces the sorted-candidate inputs, the
        # sequential capacity loop stays host-side — in native C++
        # (ops/native/greedy.cpp) at world 1, Python-object greedy with
        # all-gathered candidates at world > 1)
        if self.engine.backend == "gpu" and not spec.unlimited:
            key = (id(system), tuple(local_names))
            if self._fast_key != key:
                self._fast_sweep = FastSweep(
                    system, local_names, backend="gpu", device=self.engine.device
                )
                self._fast_key = key
            cells = self._fast_sweep.reconcile_cells()
            if cells is not None:
                stats = _ShardStats(
                    n_cells=self._fast_sweep.n_cells, n_servers=len(local_names)
                )
                initialized, _rank, world = self._dist_info()
                import os

                if (world == 1
                        and os.environ.get("INFERNO_NATIVE_GREEDY", "1") != "0"):
                    rec = self._solve_limited_native(system, local_names, cells,
                                                     spec, acc_names)
                    if rec is not None:
                        return rec, stats
                self._populate_from_cells(system, local_names, cells)
                return self._finish_slow(system, local_names, spec, acc_names, stats)
        stats = self.engine.sweep(system, server_names=local_names)
        return self._finish_slow(system, local_names, spec, acc_names, stats)

    def _greedy_static(self, system: System):
        """Static per-cell greedy inputs (units/replica, accelerator-type
        index) and per-server priorities, cached per fleet topology."""
        fs = self._fast_sweep
        cached = getattr(fs, "_greedy_static_cache", None)
        if cached is not None:
            return cached
        type_names = sorted(
            {acc.type for acc in system.accelerators.values()} | set(system.capacity)
        )
        type_index = {t: i for i, t in enumerate(type_names)}
        acc_tidx = np.empty(len(fs.acc_names), dtype=np.int32)
        for i, an in enumerate(fs.acc_names):
            acc_tidx[i] = type_index[system.accelerators[an].type]
        units = np.empty(fs.n_cells, dtype=np.int32)
        for k in range(fs.n_cells):
            srv = fs._srv_objs[fs.cell_server[k]]
            acc_name = fs.acc_names[fs.cell_acc_idx[k]]
            acc = system.accelerators[acc_name]
            model = system.models[srv.model_name]
            units[k] = model.get_num_instances(acc_name) * acc.multiplicity
        prio = np.array([s.priority(system) for s in fs._srv_objs], dtype=np.int32)
        cell_tidx = acc_tidx[fs.cell_acc_idx]
        cached = (type_names, cell_tidx, units, prio)
        fs._greedy_static_cache = cached
        return cached

    def _solve_limited_native(self, system: System, local_names, cells, spec,
                              acc_names):
        """Native limited-mode solve: vectorized candidate prep + the C++
        greedy (ops/native/greedy.cpp) + winner-only materialization.
        Returns a WinnerRecord, or None if the native library is missing."""
        from ..config import SaturationPolicy
        from ..core import Allocation

        try:
            from ..ops.sweep import run_greedy_native
        except Exception:
            return None

        type_names, cell_tidx, units, prio = self._greedy_static(system)
        fs = self._fast_sweep

        feas = cells["feasible"].astype(bool)
        idx = np.nonzero(feas)[0]
        value = cells["value"]
        srv_of = cells["cell_server"]
        # per-server candidates sorted by value; stable -> acc-name tie order
        # (identical to sorted(all_allocations.values(), key=value))
        order = idx[np.lexsort((value[idx].astype(np.float64), srv_of[idx]))]
        n_srv = len(local_names)
        counts = np.bincount(srv_of[order], minlength=n_srv)
        seg = np.zeros(n_srv + 1, dtype=np.int32)
        np.cumsum(counts, out=seg[1:])

        cand_value = np.ascontiguousarray(value[order], dtype=np.float32)
        # zero-load "" allocations drop like the Python acc-lookup miss
        cand_tidx = np.where(
            cells["zero_empty"][order].astype(bool), -1, cell_tidx[order]
        ).astype(np.int32)
        cand_units = np.ascontiguousarray(units[order], dtype=np.int32)
        cand_reps = np.ascontiguousarray(cells["num_replicas"][order], dtype=np.int32)
        capacity = np.array(
            [int(system.capacity.get(t, 0)) for t in type_names], dtype=np.int32
        )
        policy = {
            SaturationPolicy.NONE: 0,
            SaturationPolicy.PRIORITY_EXHAUSTIVE: 1,
            SaturationPolicy.PRIORITY_ROUND_ROBIN: 2,
            SaturationPolicy.ROUND_ROBIN: 3,
        }[SaturationPolicy.parse(spec.saturationPolicy)]

        try:
            win_cand, win_reps = run_greedy_native(
                cand_value, cand_tidx, cand_units, cand_reps, seg,
                np.ascontiguousarray(prio, dtype=np.int32), capacity,
                bool(spec.delayedBestEffort), policy, allow_build=False,
            )
        except Exception:
            return None

        # winner-only materialization (<= one Allocation per server)
        rec = _empty_winner(n_srv)
        acc_index = {n: i for i, n in enumerate(acc_names)}
        servers = system.servers
        for s in range(n_srv):
            servers[local_names[s]].remove_allocation()
        picked = np.nonzero(win_cand >= 0)[0]
        for s in picked.tolist():
            cell = int(order[win_cand[s]])
            acc_key = fs.acc_names[int(fs.cell_acc_idx[cell])]
            zero = bool(cells["zero_empty"][cell])
            alloc = Allocation(
                accelerator="" if zero else acc_key,
                num_replicas=int(cells["num_replicas"][cell]),
                batch_size=int(cells["batch"][cell]),
                cost=float(cells["cost"][cell]),
                value=float(cells["value"][cell]),
                itl=float(cells["itl"][cell]),
                ttft=float(cells["ttft"][cell]),
                rho=float(cells["rho"][cell]),
                max_arrv_rate_per_replica=float(cells["max_rate"][cell]),
            )
            granted = int(win_reps[s])
            if alloc.num_replicas > 0 and granted != alloc.num_replicas:
                factor = float(granted) / float(alloc.num_replicas)
                alloc.cost *= factor
                alloc.value *= factor
                alloc.num_replicas = granted
            servers[local_names[s]].set_allocation(alloc)
            rec.acc_idx[s] = -2 if alloc.accelerator == "" else acc_index[alloc.accelerator]
            rec.num_replicas[s] = alloc.num_replicas
            rec.batch[s] = alloc.batch_size
            rec.cost[s] = alloc.cost
            rec.itl[s] = alloc.itl
            rec.ttft[s] = alloc.ttft
        return rec

    def _allgather_candidates(self, system: System, local_names, acc_names, world):
        """Greedy limited mode at world>1: all-gather every rank's candidate
        allocations so each rank runs the IDENTICAL global greedy (capacity
        is a global constraint — rank-local greedy would double-spend it).
        Candidate record: [server_gidx, acc_idx, zero_empty, replicas, batch,
        cost, value, itl, ttft, rho, max_rate] (f32)."""
        import torch
        import torch.distributed as dist

        from ..core import Allocation

        all_names = sorted(system.servers)
        name_to_global = {n: i for i, n in enumerate(all_names)}
        acc_index = {n: i for i, n in enumerate(acc_names)}
        recs = []
        for name in local_names:
            gidx = float(name_to_global[name])
            for acc_key, alloc in system.servers[name].all_allocations.items():
                recs.append([
                    gidx,
                    float(acc_index[acc_key]),
                    1.0 if alloc.accelerator == "" else 0.0,
                    float(alloc.num_replicas),
                    float(alloc.batch_size),
                    alloc.cost,
                    alloc.value,
                    alloc.itl,
                    alloc.ttft,
                    alloc.rho,
                    alloc.max_arrv_rate_per_replica,
                ])
        local = torch.tensor(recs, dtype=torch.float32).reshape(-1, 11)
        backend = dist.get_backend(self.group)
        comm_dev = "cuda" if backend == "nccl" else "cpu"
        local = local.to(comm_dev)
        # variable-size all_gather: exchange lengths first
        n_local = torch.tensor([local.shape[0]], dtype=torch.int64, device=comm_dev)
        sizes = [torch.zeros_like(n_local) for _ in range(world)]
        dist.all_gather(sizes, n_local, group=self.group)
        max_n = int(max(s.item() for s in sizes))
        padded = torch.zeros((max_n, 11), dtype=torch.float32, device=comm_dev)
        padded[: local.shape[0]] = local
        gathered = [torch.empty_like(padded) for _ in range(world)]
        dist.all_gather(gathered, padded, group=self.group)
        # rebuild every server's candidate map (identical on all ranks)
        for name in all_names:
            system.servers[name].all_allocations = {}
        for r, g in enumerate(gathered):
            rows = g[: int(sizes[r].item())].cpu().numpy()
            for row in rows:
                name = all_names[int(row[0])]
                acc_key = acc_names[int(row[1])]
                system.servers[name].all_allocations[acc_key] = Allocation(
                    accelerator="" if row[2] > 0.5 else acc_key,
                    num_replicas=int(row[3]),
                    batch_size=int(row[4]),
                    cost=float(row[5]),
                    value=float(row[6]),
                    itl=float(row[7]),
                    ttft=float(row[8]),
                    rho=float(row[9]),
                    max_arrv_rate_per_replica=float(row[10]),
                )

    def _finish_slow(self, system: System, local_names, spec: OptimizerSpec, acc_names,
                     stats):
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
            # greedy limited mode needs the full candidate lists and capacity
            # is a GLOBAL constraint: at world>1 each rank all-gathers every
            # shard's candidates and runs the identical global greedy
            from ..solver.greedy import solve_greedy

            initialized, _rank, world = self._dist_info()
            if initialized and world > 1:
                self._allgather_candidates(system, local_names, acc_names, world)
            solve_greedy(
                system,
                delayed_best_effort=spec.delayedBestEffort,
                saturation_policy=SaturationPolicy.parse(spec.saturationPolicy),
            )

        acc_index = {n: i for i, n in enumerate(acc_names)}
        rec = _empty_winner(len(local_names))
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
