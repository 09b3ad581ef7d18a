"""FastSweep — the production reconcile hot path.

The full snapshot builder (snapshot.py) loops per cell in Python; at fleet
scale that host time dominates the reconcile (the HIP sweep itself is ~ms).
FastSweep splits the snapshot into:

  * a STATIC cell structure built once per fleet topology (perf parms,
    SLO targets, cell->server segments, accelerator costs), and
  * a vectorized DYNAMIC refresh per reconcile (arrival rates, token
    averages, current-allocation penalty inputs) — pure numpy gathers.

Per reconcile it uploads the refreshed SoA, launches wva_sweep + wva_argmin,
and materializes ONLY the per-server winner records (not all candidate
cells). The slow path (SweepEngine) remains the oracle and serves greedy
limited mode, which needs the full candidate lists.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

from ..config import MAX_BATCH_STATES
from ..core import create_allocation
from ..core.system import System
from .snapshot import FLAG_CUR_EMPTY, FLAG_CUR_SAME, FLAG_HAS_CUR

_INT_KEYS = ("in_tok", "out_tok", "batch_n", "min_replicas", "perf_max_batch",
             "cur_replicas", "flags")
_FLOAT_KEYS = ("alpha", "beta", "gamma", "delta", "arrival_rate", "t_itl", "t_ttft",
               "t_tps", "acc_cost", "cur_cost")


@dataclass
class WinnerRecord:
    """Winner allocation per server (arrays indexed by segment)."""

    acc_idx: np.ndarray  # int32, -1 = none, -2 = zero-load empty
    num_replicas: np.ndarray  # int32
    batch: np.ndarray  # int32
    cost: np.ndarray  # float32
    value: np.ndarray
    itl: np.ndarray
    ttft: np.ndarray
    rho: np.ndarray
    max_rate: np.ndarray


class FastSweep:
    def __init__(self, system: System, server_names: Optional[list[str]] = None,
                 backend: str = "gpu", device: str = "cuda"):
        self.system = system
        self.backend = backend
        self.device = device
        self.server_names = (
            list(server_names) if server_names is not None else sorted(system.servers)
        )
        self.acc_names = sorted(system.accelerators)
        self._acc_index = {n: i for i, n in enumerate(self.acc_names)}
        self._build_static()

    # ------------------------------------------------------------------
    def _build_static(self) -> None:
        system = self.system
        cell_server: list[int] = []
        cell_acc_idx: list[int] = []
        stat = {k: [] for k in ("alpha", "beta", "gamma", "delta", "acc_cost",
                                "t_itl", "t_ttft", "t_tps",
                                "perf_max_batch_cfg", "at_tokens", "override",
                                "min_replicas")}
        seg_start = [0]
        self._srv_objs = []
        for seg, name in enumerate(self.server_names):
            server = system.servers[name]
            self._srv_objs.append(server)
            model = system.models.get(server.model_name)
            svc = system.service_classes.get(server.service_class_name)
            target = svc.model_target(server.model_name) if svc is not None else None
            if model is not None and target is not None:
                for acc_name in sorted(server.candidate_accelerators(system)):
                    acc = system.accelerators[acc_name]
                    perf = model.get_perf_data(acc_name)
                    if perf is None:
                        continue
                    cell_server.append(seg)
                    cell_acc_idx.append(self._acc_index[acc_name])
                    stat["alpha"].append(perf.decodeParms.alpha)
                    stat["beta"].append(perf.decodeParms.beta)
                    stat["gamma"].append(perf.prefillParms.gamma)
                    stat["delta"].append(perf.prefillParms.delta)
                    stat["acc_cost"].append(acc.cost * model.get_num_instances(acc_name))
                    stat["t_itl"].append(target.itl)
                    stat["t_ttft"].append(target.ttft)
                    stat["t_tps"].append(target.tps)
                    stat["perf_max_batch_cfg"].append(perf.maxBatchSize)
                    stat["at_tokens"].append(perf.atTokens)
                    stat["override"].append(server.max_batch_size)
                    stat["min_replicas"].append(server.min_num_replicas)
            seg_start.append(len(cell_server))

        self.n_cells = len(cell_server)
        self.cell_server = np.asarray(cell_server, dtype=np.int64)
        self.cell_acc_idx = np.asarray(cell_acc_idx, dtype=np.int32)
        self.seg_start = np.asarray(seg_start, dtype=np.int32)
        self._stat = {
            k: np.asarray(v, dtype=np.float32 if k in
                          ("alpha", "beta", "gamma", "delta", "acc_cost",
                           "t_itl", "t_ttft", "t_tps") else np.int32)
            for k, v in stat.items()
        }
        self._torch_cache = None

    # ------------------------------------------------------------------
    def _refresh_dynamic(self) -> dict:
        """Gather per-server dynamic state into per-cell arrays (vectorized)."""
        n_srv = len(self.server_names)
        arrival = np.empty(n_srv, dtype=np.float32)
        in_tok = np.empty(n_srv, dtype=np.int32)
        out_tok = np.empty(n_srv, dtype=np.int32)
        cur_acc = np.empty(n_srv, dtype=np.int32)  # -1 empty, -3 no cur
        cur_rep = np.zeros(n_srv, dtype=np.int32)
        cur_cost = np.zeros(n_srv, dtype=np.float32)
        for i, server in enumerate(self._srv_objs):
            load = server.load
            arrival[i] = load.arrivalRate if load is not None else 0.0
            in_tok[i] = load.avgInTokens if load is not None else 0
            out_tok[i] = load.avgOutTokens if load is not None else 0
            cur = server.cur_allocation
            if cur is None:
                cur_acc[i] = -3
            else:
                cur_acc[i] = self._acc_index.get(cur.accelerator, -1)
                cur_rep[i] = cur.num_replicas
                cur_cost[i] = cur.cost

        cs = self.cell_server
        c_out = out_tok[cs]
        safe_k = np.maximum(c_out, 1)
        n_scaled = np.maximum(
            (self._stat["perf_max_batch_cfg"].astype(np.int64)
             * self._stat["at_tokens"].astype(np.int64)) // safe_k,
            1,
        )
        batch_n = np.where(self._stat["override"] > 0, self._stat["override"], n_scaled)
        batch_n = np.minimum(batch_n, MAX_BATCH_STATES).astype(np.int32)
        # zero-load cells take the in-kernel zero path; batch_n is unused
        # there but must stay small so it doesn't inflate the LDS budget
        zero_load = (arrival[cs] == 0) | (c_out == 0)
        batch_n = np.where(zero_load, 1, batch_n).astype(np.int32)
        pmb = np.where(
            self._stat["override"] > 0, self._stat["override"],
            self._stat["perf_max_batch_cfg"],
        ).astype(np.int32)

        c_cur_acc = cur_acc[cs]
        flags = np.where(c_cur_acc != -3, FLAG_HAS_CUR, 0).astype(np.int32)
        flags |= np.where(c_cur_acc == self.cell_acc_idx, FLAG_CUR_SAME, 0)
        flags |= np.where(c_cur_acc == -1, FLAG_CUR_EMPTY, 0)

        return {
            "in_tok": in_tok[cs],
            "out_tok": c_out,
            "batch_n": batch_n,
            "min_replicas": self._stat["min_replicas"],
            "perf_max_batch": pmb,
            "cur_replicas": cur_rep[cs],
            "flags": flags,
            "alpha": self._stat["alpha"],
            "beta": self._stat["beta"],
            "gamma": self._stat["gamma"],
            "delta": self._stat["delta"],
            "arrival_rate": arrival[cs],
            "t_itl": self._stat["t_itl"],
            "t_ttft": self._stat["t_ttft"],
            "t_tps": self._stat["t_tps"],
            "acc_cost": self._stat["acc_cost"],
            "cur_cost": cur_cost[cs],
        }

    def cell_arrays(self) -> dict:
        """Torch CPU tensors of the refreshed snapshot (test/inspection API)."""
        import torch

        arrs = self._refresh_dynamic()
        out = {}
        for k in _INT_KEYS:
            out[k] = torch.from_numpy(np.ascontiguousarray(arrs[k], dtype=np.int32))
        for k in _FLOAT_KEYS:
            out[k] = torch.from_numpy(np.ascontiguousarray(arrs[k], dtype=np.float32))
        return out

    # ------------------------------------------------------------------
    def reconcile(self) -> WinnerRecord:
        if self.backend == "gpu":
            return self._reconcile_gpu()
        return self._reconcile_cpu()

    def _reconcile_gpu(self) -> WinnerRecord:
        import os
        import time

        import torch

        from ..ops.sweep import run_argmin, run_sweep

        n_srv = len(self.server_names)
        if self.n_cells == 0:
            return _empty_winner(n_srv)
        trace = os.environ.get("INFERNO_TIMING", "") == "1"
        if trace:
            torch.cuda.synchronize()
            t0 = time.perf_counter()
        arrays = self.cell_arrays()
        if trace:
            t1 = time.perf_counter()
        out = run_sweep(arrays, device=self.device)
        seg = torch.from_numpy(self.seg_start).to(self.device)
        winner = run_argmin(out.value, out.feasible, seg)
        if trace:
            torch.cuda.synchronize()
            t2 = time.perf_counter()
            print(
                f"[timing] refresh={1e3*(t1-t0):.2f}ms sweep+argmin={1e3*(t2-t1):.2f}ms",
                flush=True,
            )
        # single D2H sync for everything needed to materialize winners
        w = winner.cpu().numpy()
        stack = torch.stack(
            [
                out.cost,
                out.value,
                out.itl,
                out.ttft,
                out.rho,
                out.max_rate,
            ]
        ).cpu().numpy()
        reps = out.num_replicas.cpu().numpy()
        batch = out.batch.cpu().numpy()
        zero_empty = out.zero_empty.cpu().numpy()

        has = w >= 0
        wc = np.where(has, w, 0)
        acc_idx = np.where(
            has, np.where(zero_empty[wc] > 0, -2, self.cell_acc_idx[wc]), -1
        ).astype(np.int32)
        return WinnerRecord(
            acc_idx=acc_idx,
            num_replicas=np.where(has, reps[wc], 0).astype(np.int32),
            batch=np.where(has, batch[wc], 0).astype(np.int32),
            cost=np.where(has, stack[0][wc], 0.0).astype(np.float32),
            value=np.where(has, stack[1][wc], 0.0).astype(np.float32),
            itl=np.where(has, stack[2][wc], 0.0).astype(np.float32),
            ttft=np.where(has, stack[3][wc], 0.0).astype(np.float32),
            rho=np.where(has, stack[4][wc], 0.0).astype(np.float32),
            max_rate=np.where(has, stack[5][wc], 0.0).astype(np.float32),
        )

    def _reconcile_cpu(self) -> WinnerRecord:
        """Golden scalar path returning the same winner records."""
        n_srv = len(self.server_names)
        rec = _empty_winner(n_srv)
        for seg, name in enumerate(self.server_names):
            server = self.system.servers[name]
            best = None
            for acc_name in sorted(server.candidate_accelerators(self.system)):
                alloc = create_allocation(self.system, name, acc_name)
                if alloc is None:
                    continue
                if server.cur_allocation is not None:
                    alloc.value = server.cur_allocation.transition_penalty(alloc)
                if best is None or alloc.value < best.value:
                    best = alloc
            if best is None:
                continue
            rec.acc_idx[seg] = (
                -2 if best.accelerator == "" else self._acc_index[best.accelerator]
            )
            rec.num_replicas[seg] = best.num_replicas
            rec.batch[seg] = best.batch_size
            rec.cost[seg] = best.cost
            rec.value[seg] = best.value
            rec.itl[seg] = best.itl
            rec.ttft[seg] = best.ttft
            rec.rho[seg] = best.rho
            rec.max_rate[seg] = best.max_arrv_rate_per_replica
        return rec


def _empty_winner(n: int) -> WinnerRecord:
    return WinnerRecord(
        acc_idx=np.full(n, -1, dtype=np.int32),
        num_replicas=np.zeros(n, dtype=np.int32),
        batch=np.zeros(n, dtype=np.int32),
        cost=np.zeros(n, dtype=np.float32),
        value=np.zeros(n, dtype=np.float32),
        itl=np.zeros(n, dtype=np.float32),
        ttft=np.zeros(n, dtype=np.float32),
        rho=np.zeros(n, dtype=np.float32),
        max_rate=np.zeros(n, dtype=np.float32),
    )
