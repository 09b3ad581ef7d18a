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

    # ------------------------------------------------------------------
    # Array overrides for the steady-state loop: when set, the per-server
    # Python object reads are skipped entirely (the bench/controller can
    # maintain these arrays from winner records between reconciles).
    # load_override = (arrival f32, in_tok i32, out_tok i32) per local server;
    # cur_override = (cur_acc i32 [-3 none, -1 empty, >=0 acc index],
    #                 cur_replicas i32, cur_cost f32).
    load_override = None
    cur_override = None

    def _refresh_dynamic(self) -> dict:
        """Gather per-server dynamic state into per-cell arrays (vectorized)."""
        n_srv = len(self.server_names)
        if self.load_override is not None:
            arrival, in_tok, out_tok = self.load_override
            arrival = np.asarray(arrival, dtype=np.float32)
            in_tok = np.asarray(in_tok, dtype=np.int32)
            out_tok = np.asarray(out_tok, dtype=np.int32)
        else:
            arrival = np.empty(n_srv, dtype=np.float32)
            in_tok = np.empty(n_srv, dtype=np.int32)
            out_tok = np.empty(n_srv, dtype=np.int32)
            for i, server in enumerate(self._srv_objs):
                load = server.load
                arrival[i] = load.arrivalRate if load is not None else 0.0
                in_tok[i] = load.avgInTokens if load is not None else 0
                out_tok[i] = load.avgOutTokens if load is not None else 0
        if self.cur_override is not None:
            cur_acc, cur_rep, cur_cost = self.cur_override
            cur_acc = np.asarray(cur_acc, dtype=np.int32)
            cur_rep = np.asarray(cur_rep, dtype=np.int32)
            cur_cost = np.asarray(cur_cost, dtype=np.float32)
        else:
            cur_acc = np.empty(n_srv, dtype=np.int32)  # -1 empty, -3 no cur
            cur_rep = np.zeros(n_srv, dtype=np.int32)
            cur_cost = np.zeros(n_srv, dtype=np.float32)
            for i, server in enumerate(self._srv_objs):
                cur = server.cur_allocation
                if cur is None:
                    cur_acc[i] = -3
                else:
                    cur_acc[i] = self._acc_index.get(cur.accelerator, -1)
                    cur_rep[i] = cur.num_replicas
                    cur_cost[i] = cur.cost

        cs = self.cell_server
        c_out = out_tok[cs]
        # batch sizing depends only on out_tok (and static profile fields);
        # steady-state fleets keep token averages stable, so cache it
        key = out_tok.tobytes()
        cached = getattr(self, "_batch_cache", None)
        if cached is not None and cached[0] == key:
            base_batch, pmb = cached[1], cached[2]
        else:
            safe_k = np.maximum(c_out, 1)
            n_scaled = np.maximum(
                (self._stat["perf_max_batch_cfg"].astype(np.int64)
                 * self._stat["at_tokens"].astype(np.int64)) // safe_k,
                1,
            )
            # uncapped like the reference (allocation.go:80-86); huge-N cells
            # dispatch to the GMEM spill bucket in choose_buckets
            base_batch = np.where(
                self._stat["override"] > 0, self._stat["override"], n_scaled
            ).astype(np.int32)
            pmb = np.where(
                self._stat["override"] > 0, self._stat["override"],
                self._stat["perf_max_batch_cfg"],
            ).astype(np.int32)
            self._batch_cache = (key, base_batch, pmb)
        # zero-load cells take the in-kernel zero path; batch_n is unused
        # there but must stay small so it doesn't inflate the LDS budget
        zero_load = (arrival[cs] == 0) | (c_out == 0)
        batch_n = np.where(zero_load, 1, base_batch).astype(np.int32)

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

    # GPU persistent-state reconcile: static SoA uploaded once; per step the
    # dynamic fields are packed into TWO pinned staging buffers (one int32,
    # one fp32) -> two H2D copies; bucket ids cached on the batch_n bytes;
    # outputs preallocated; winner records gathered ON-DEVICE and downloaded
    # in a single D2H copy.
    _DYN_INT = ("in_tok", "out_tok", "batch_n", "perf_max_batch", "cur_replicas", "flags")
    _STAT_F32 = ("alpha", "beta", "gamma", "delta", "t_itl", "t_ttft", "t_tps", "acc_cost")

    def _init_gpu_state(self) -> None:
        import torch

        from ..ops.sweep import load_library

        load_library(allow_build=False)
        dev = self.device
        n = self.n_cells
        st: dict = {}
        for k in self._STAT_F32:
            st[k] = torch.from_numpy(np.ascontiguousarray(self._stat[k])).to(dev)
        st["min_replicas"] = torch.from_numpy(
            np.ascontiguousarray(self._stat["min_replicas"], dtype=np.int32)
        ).to(dev)
        st["pin_i"] = torch.empty((len(self._DYN_INT), n), dtype=torch.int32,
                                  pin_memory=True)
        st["pin_f"] = torch.empty((2, n), dtype=torch.float32, pin_memory=True)
        st["dev_i"] = torch.empty((len(self._DYN_INT), n), dtype=torch.int32, device=dev)
        st["dev_f"] = torch.empty((2, n), dtype=torch.float32, device=dev)
        st["seg"] = torch.from_numpy(self.seg_start).to(dev)
        st["cell_acc"] = torch.from_numpy(self.cell_acc_idx).to(dev)
        # outputs
        st["feasible"] = torch.zeros(n, dtype=torch.uint8, device=dev)
        st["zero_empty"] = torch.zeros(n, dtype=torch.uint8, device=dev)
        st["num_replicas"] = torch.zeros(n, dtype=torch.int32, device=dev)
        st["batch"] = torch.zeros(n, dtype=torch.int32, device=dev)
        for k in ("cost", "value", "itl", "ttft", "rho", "max_rate"):
            st[k] = torch.zeros(n, dtype=torch.float32, device=dev)
        n_srv = len(self.server_names)
        st["winner"] = torch.full((n_srv,), -1, dtype=torch.int32, device=dev)
        st["gather_f"] = torch.empty((6, n_srv), dtype=torch.float32, device=dev)
        st["gather_i"] = torch.empty((4, n_srv), dtype=torch.int32, device=dev)
        st["pin_out_f"] = torch.empty((6, n_srv), dtype=torch.float32, pin_memory=True)
        st["pin_out_i"] = torch.empty((4, n_srv), dtype=torch.int32, pin_memory=True)
        st["bucket_key"] = None
        st["buckets"] = []
        self._gpu = st

    def _buckets_for(self, batch_n: np.ndarray):
        import torch

        from ..ops.sweep import alloc_gmem_slabs, choose_buckets

        st = self._gpu
        key = batch_n.tobytes()
        if st["bucket_key"] == key:
            return st["buckets"]
        buckets = []
        for nt, ids, bmax, count, gmem in choose_buckets(batch_n):
            ids_t = torch.from_numpy(ids).to(self.device) if ids is not None else None
            slabs = (
                alloc_gmem_slabs(count, bmax, nt, self.device) if gmem else (None, None)
            )
            buckets.append((nt, ids_t, bmax, count, slabs))
        st["bucket_key"] = key
        st["buckets"] = buckets
        return buckets

    def _ensure_ctx(self) -> None:
        """Create the native reconcile context (opaque C struct owning the
        HIP streams/events and all registered device/pinned pointers) —
        the whole per-tick pipeline then runs behind ONE C call."""
        import ctypes

        from ..ops.sweep import load_library

        st = self._gpu
        if "ctx" in st:
            return
        lib = load_library(allow_build=False)
        di, df = st["dev_i"], st["dev_f"]
        slots = [
            di, df,
            st["min_replicas"],
            st["alpha"], st["beta"], st["gamma"], st["delta"],
            st["t_itl"], st["t_ttft"], st["t_tps"], st["acc_cost"],
            st["feasible"], st["zero_empty"], st["num_replicas"], st["batch"],
            st["cost"], st["value"], st["itl"], st["ttft"], st["rho"],
            st["max_rate"],
            st["seg"], st["winner"], st["cell_acc"], st["gather_f"],
            st["gather_i"],
            st["pin_i"], st["pin_f"], st["pin_out_f"], st["pin_out_i"],
        ]
        arr = (ctypes.c_void_p * len(slots))(
            *[ctypes.c_void_p(t.data_ptr()) for t in slots]
        )
        ctx = lib.wva_ctx_create(
            ctypes.c_int(self.n_cells), ctypes.c_int(len(self.server_names)), arr
        )
        if not ctx:
            from ..ops.sweep import HipKernelError

            raise HipKernelError("wva_ctx_create failed")
        st["ctx"] = ctx
        st["ctx_lib"] = lib

    def __del__(self):  # release the native context's streams/events
        st = getattr(self, "_gpu", None)
        if st and "ctx" in st:
            try:
                st["ctx_lib"].wva_ctx_destroy(st.pop("ctx"))
            except Exception:
                pass

    def _sync_buckets(self, batch_n: np.ndarray) -> None:
        """Push the bucket layout + analyzer mode into the native context
        (only when it changed)."""
        import ctypes

        st = self._gpu
        buckets = self._buckets_for(batch_n)
        if st.get("ctx_buckets_key") == st["bucket_key"]:
            return
        n = len(buckets)
        nts = (ctypes.c_int * n)(*[b[0] for b in buckets])
        ids = (ctypes.c_void_p * n)(
            *[ctypes.c_void_p(b[1].data_ptr()) if b[1] is not None else None
              for b in buckets]
        )
        nbl = (ctypes.c_int * n)(
            *[int(b[1].numel()) if b[1] is not None else self.n_cells for b in buckets]
        )
        mxn = (ctypes.c_int * n)(*[b[2] for b in buckets])
        g_invs = (ctypes.c_void_p * n)(
            *[ctypes.c_void_p(b[4][0].data_ptr()) if b[4][0] is not None else None
              for b in buckets]
        )
        g_anchors = (ctypes.c_void_p * n)(
            *[ctypes.c_void_p(b[4][1].data_ptr()) if b[4][1] is not None else None
              for b in buckets]
        )
        mode = 1 if getattr(self.system, "analyzer_mode", "mm1k") == "mg1" else 0
        cv2 = float(getattr(self.system, "analyzer_cv2", 1.0))
        rc = st["ctx_lib"].wva_ctx_set_buckets(
            st["ctx"], ctypes.c_int(n), nts, ids, nbl, mxn,
            ctypes.c_int(mode), ctypes.c_float(cv2), g_invs, g_anchors,
        )
        if rc != 0:
            from ..ops.sweep import HipKernelError

            raise HipKernelError(f"wva_ctx_set_buckets failed: {rc}")
        st["ctx_buckets_key"] = st["bucket_key"]

    def _fill_pinned(self, arrs: dict) -> None:
        import torch

        st = self._gpu
        pin_i, pin_f = st["pin_i"], st["pin_f"]
        for j, k in enumerate(self._DYN_INT):
            pin_i[j] = torch.from_numpy(np.ascontiguousarray(arrs[k], dtype=np.int32))
        pin_f[0] = torch.from_numpy(
            np.ascontiguousarray(arrs["arrival_rate"], dtype=np.float32)
        )
        pin_f[1] = torch.from_numpy(np.ascontiguousarray(arrs["cur_cost"], dtype=np.float32))

    def _native_reconcile(self, arrs: dict) -> None:
        from ..ops.sweep import HipKernelError

        st = self._gpu
        self._ensure_ctx()
        self._fill_pinned(arrs)
        self._sync_buckets(arrs["batch_n"])
        rc = st["ctx_lib"].wva_reconcile(st["ctx"])
        if rc != 0:
            raise HipKernelError(f"wva_reconcile failed: {rc}")

    def reconcile_cells(self) -> Optional[dict]:
        """Run the sweep and return PER-CELL output arrays (numpy) plus the
        cell metadata — the input the greedy limited-mode solver needs (full
        candidate lists, not just argmin winners). GPU backend only; returns
        None on CPU (the slow engine path covers it)."""
        if self.backend != "gpu" or self.n_cells == 0:
            return None
        if not hasattr(self, "_gpu"):
            self._init_gpu_state()
        st = self._gpu
        arrs = self._refresh_dynamic()
        self._native_reconcile(arrs)  # fully synced on return
        out = {
            k: st[k].cpu().numpy()
            for k in ("feasible", "zero_empty", "num_replicas", "batch", "cost",
                      "value", "itl", "ttft", "rho", "max_rate")
        }
        out["cell_server"] = self.cell_server
        out["cell_acc_idx"] = self.cell_acc_idx
        out["seg_start"] = self.seg_start
        return out

    def _reconcile_gpu(self) -> WinnerRecord:
        n_srv = len(self.server_names)
        if self.n_cells == 0:
            return _empty_winner(n_srv)
        if not hasattr(self, "_gpu"):
            self._init_gpu_state()
        st = self._gpu
        arrs = self._refresh_dynamic()
        self._native_reconcile(arrs)
        of = st["pin_out_f"].numpy()
        oi = st["pin_out_i"].numpy()
        return WinnerRecord(
            acc_idx=oi[0].copy(),
            num_replicas=oi[1].copy(),
            batch=oi[2].copy(),
            cost=of[0].copy(),
            value=of[1].copy(),
            itl=of[2].copy(),
            ttft=of[3].copy(),
            rho=of[4].copy(),
            max_rate=of[5].copy(),
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
