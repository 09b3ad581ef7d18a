"""ctypes driver for the HIP allocate-sweep kernels.

Loads the in-tree ``lib/libwva_hip.so`` and launches on torch's current HIP
stream with raw device pointers. On a GPU box a missing/failed native library
raises loudly (no silent eager fallback) — the CPU path must be requested
explicitly via the engine's ``backend="cpu"``.
"""
from __future__ import annotations

import ctypes
import os
from dataclasses import dataclass
from typing import Optional

from .build import LIB_PATH, build, needs_build

MAX_N = 8192  # must match WVA_MAX_N in wva_kernels.hip (64KB-LDS limit)
# XL LDS tier: fits CDNA4's 160 KiB LDS with the >64KB dynamic-LDS opt-in
# (must match WVA_XL_MAX_N); override to MAX_N to force such cells to GMEM
XL_MAX_N = int(os.environ.get("INFERNO_XL_MAX_N", "32768"))
HUGE_MAX_N = 1 << 22  # must match WVA_HUGE_MAX_N (global-memory spill limit)
# N-bucket thresholds (must match WVA_N_SMALL / WVA_N_MED): cells are
# dispatched to 64- and 256-thread blocks by batch size — widths picked by
# A/B measurement on MI355X (see choose_buckets); small/medium cells run one
# barrier-free wave each, large-N cells get 4-wave blocks.
N_SMALL = int(os.environ.get("INFERNO_N_SMALL", "512"))
N_MED = int(os.environ.get("INFERNO_N_MED", "2048"))


def choose_buckets(batch_n):
    """Partition cells into N-buckets and pick each bucket's block size.

    The natural guess — wide blocks shorten a single cell's latency — is
    WRONG on CDNA4 for this kernel (measured, profiles/REPORT.md): per-eval
    cost is dominated by the redundant per-lane work, which costs
    waves-per-SIMD x instructions, while the serial chain states are cheap
    fp32-LDS FMAs. Narrow blocks therefore win in BOTH regimes.

    Cells with MAX_N < N <= XL_MAX_N use the XL tier: still LDS-resident via
    the 160 KiB dynamic-LDS opt-in (one workgroup per CU, fine for the rare
    huge cells). Cells above XL_MAX_N go to the global-memory bucket: their
    chain geometry lives in a per-block HBM slab instead of LDS (GMEM kernel
    instantiation). Either way the sweep is uncapped like the reference
    (allocation.go:80-86).

    Returns [(nt, cell_idx int32 array or None, bucket_max_n, count, gmem)].
    """
    import numpy as np

    n = len(batch_n)
    masks = [
        (64, batch_n <= N_SMALL),
        (256, (batch_n > N_SMALL) & (batch_n <= N_MED)),
        (1024, (batch_n > N_MED) & (batch_n <= MAX_N)),
        ("xl", (batch_n > MAX_N) & (batch_n <= XL_MAX_N)),
        ("huge", batch_n > XL_MAX_N),
    ]
    out = []
    for base_nt, mask in masks:
        idx = np.nonzero(mask)[0]
        count = len(idx)
        if count == 0:
            continue
        if base_nt == "xl":
            nt = int(os.environ.get("INFERNO_NT_XL", "256"))
            ids = None if count == n else idx.astype(np.int32)
            out.append((nt, ids, int(batch_n[mask].max()), count, False))
            continue
        if base_nt == "huge":
            bmax = int(batch_n[mask].max())
            if bmax > HUGE_MAX_N:
                raise HipKernelError(
                    f"batch size {bmax} exceeds the global-memory sweep limit "
                    f"{HUGE_MAX_N} (per-cell geometry would be "
                    f"{bmax * 4 / 1e6:.0f}+ MB)"
                )
            nt = int(os.environ.get("INFERNO_NT_HUGE", "256"))
            ids = None if count == n else idx.astype(np.int32)
            out.append((nt, ids, bmax, count, True))
            continue
        # Measured on MI355X (profiles/REPORT.md, 512-model fleet A/B): the
        # per-eval cost of a bisection step is dominated by the REDUNDANT
        # per-lane work (log/exp/expm1 of the tail, loop headers) which costs
        # waves-per-SIMD x instructions — 16-wave blocks pay 4x what 4-wave
        # blocks pay — while the serial per-lane chain states are cheap fp32
        # LDS FMAs. So each bucket wants the NARROWEST block whose serial
        # states/lane stay modest (~64): N<=512 -> 64 threads, N<=2048 -> 64
        # (66 states/lane), N<=8192 -> 256 (66 states/lane). 1024-thread
        # blocks lost 25-43% end-to-end in the A/B (ab_large_*/ab2_* runs).
        nt = {64: 64, 256: 64, 1024: 256}[base_nt]
        # measurement override (A/B tuning): INFERNO_NT_SMALL/_MED/_LARGE
        env = os.environ.get(
            {64: "INFERNO_NT_SMALL", 256: "INFERNO_NT_MED", 1024: "INFERNO_NT_LARGE"}[base_nt]
        )
        if env:
            nt = int(env)
        ids = None if count == n else idx.astype(np.int32)
        out.append((nt, ids, int(batch_n[mask].max()), count, False))
    # launch order = descending max_n (most expensive bucket first): ROCm
    # multiplexes streams onto GPU_MAX_HW_QUEUES (default 4) hardware queues,
    # so with 5 buckets the later streams share a queue and serialize —
    # measured: the XL bucket queued behind the large bucket, costing ~250us
    # of lost overlap per reconcile. Put the stragglers on distinct queues
    # and let the cheap buckets share.
    out.sort(key=lambda b: -b[2])
    return out


def alloc_gmem_slabs(n_blocks: int, max_n: int, nt: int, device: str):
    """Allocate the global-memory geometry slabs for a huge-N bucket:
    per-block chunk-transposed fp32 reciprocals + fp64 anchor log-prefixes
    (the same layout the LDS path uses)."""
    import torch

    chunk = (max_n + nt - 1) // nt
    ksub = (chunk + 31) // 32
    g_inv = torch.empty((n_blocks, chunk * nt), dtype=torch.float32, device=device)
    g_anchor = torch.empty((n_blocks, nt * ksub), dtype=torch.float64, device=device)
    return g_inv, g_anchor

_lib: Optional[ctypes.CDLL] = None


class HipKernelError(RuntimeError):
    pass


def load_library(allow_build: bool = True) -> ctypes.CDLL:
    """dlopen the kernel library (building it first if sources are newer)."""
    global _lib
    if _lib is not None:
        return _lib
    if needs_build():
        if not allow_build:
            raise HipKernelError(
                f"HIP kernel library missing or stale at {LIB_PATH}; "
                "run python -m inferno_amd.ops.build"
            )
        build()
    lib = ctypes.CDLL(LIB_PATH)
    lib.wva_sweep_launch.restype = ctypes.c_int
    lib.wva_sweep_launch_bucket.restype = ctypes.c_int
    lib.wva_argmin_launch.restype = ctypes.c_int
    lib.wva_device_count.restype = ctypes.c_int
    lib.wva_ctx_create.restype = ctypes.c_void_p
    lib.wva_ctx_create.argtypes = [ctypes.c_int, ctypes.c_int,
                                   ctypes.POINTER(ctypes.c_void_p)]
    lib.wva_ctx_set_buckets.restype = ctypes.c_int
    lib.wva_ctx_set_buckets.argtypes = [
        ctypes.c_void_p,
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_int),
        ctypes.POINTER(ctypes.c_void_p),
        ctypes.POINTER(ctypes.c_int),
        ctypes.POINTER(ctypes.c_int),
        ctypes.c_int,
        ctypes.c_float,
        ctypes.POINTER(ctypes.c_void_p),  # per-bucket g_inv slabs (huge-N)
        ctypes.POINTER(ctypes.c_void_p),  # per-bucket g_anchor slabs
    ]
    lib.wva_reconcile.restype = ctypes.c_int
    lib.wva_reconcile.argtypes = [ctypes.c_void_p]
    lib.wva_greedy_solve.restype = ctypes.c_int
    lib.wva_ctx_destroy.restype = None
    lib.wva_ctx_destroy.argtypes = [ctypes.c_void_p]
    _lib = lib
    return lib


def _ptr(t):
    return ctypes.c_void_p(t.data_ptr())


@dataclass
class SweepOutputs:
    feasible: "object"  # torch uint8 [cells]
    zero_empty: "object"
    num_replicas: "object"  # int32
    batch: "object"  # int32
    cost: "object"  # float32
    value: "object"
    itl: "object"
    ttft: "object"
    rho: "object"
    max_rate: "object"


def run_sweep(arrays: dict, device: str = "cuda", analyzer_mode: int = 0,
              cv2: float = 1.0) -> SweepOutputs:
    """Launch the sweep kernel over cell SoA arrays (torch CPU tensors in,
    results copied back to CPU tensors).

    ``arrays`` keys (from engine.snapshot.build_cell_arrays): int32 tensors
    in_tok,out_tok,batch_n,min_replicas,perf_max_batch,cur_replicas,flags and
    float32 tensors alpha,beta,gamma,delta,arrival_rate,t_itl,t_ttft,t_tps,
    acc_cost,cur_cost.
    """
    import torch

    if not torch.cuda.is_available():
        raise HipKernelError("run_sweep requires a GPU (use the CPU engine backend instead)")
    lib = load_library()
    n_cells = int(arrays["in_tok"].shape[0])

    dev = {k: v.to(device, non_blocking=True) for k, v in arrays.items()}
    t = torch
    out = SweepOutputs(
        feasible=t.zeros(n_cells, dtype=t.uint8, device=device),
        zero_empty=t.zeros(n_cells, dtype=t.uint8, device=device),
        num_replicas=t.zeros(n_cells, dtype=t.int32, device=device),
        batch=t.zeros(n_cells, dtype=t.int32, device=device),
        cost=t.zeros(n_cells, dtype=t.float32, device=device),
        value=t.zeros(n_cells, dtype=t.float32, device=device),
        itl=t.zeros(n_cells, dtype=t.float32, device=device),
        ttft=t.zeros(n_cells, dtype=t.float32, device=device),
        rho=t.zeros(n_cells, dtype=t.float32, device=device),
        max_rate=t.zeros(n_cells, dtype=t.float32, device=device),
    )
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)

    # partition cells into regime-adaptive N-buckets (+ huge-N GMEM spill)
    buckets = []
    for nt, ids, bmax, count, gmem in choose_buckets(arrays["batch_n"].numpy()):
        ids_t = torch.from_numpy(ids).to(device) if ids is not None else None
        slabs = alloc_gmem_slabs(count, bmax, nt, device) if gmem else (None, None)
        buckets.append((nt, ids_t, bmax, slabs))

    # overlap the bucket launches on separate HIP streams: wall time becomes
    # the straggler bucket's latency instead of the sum of all three
    main_stream = torch.cuda.current_stream()
    side = [torch.cuda.Stream() for _ in range(max(len(buckets) - 1, 0))]
    for i, (nt, ids, bmax, slabs) in enumerate(buckets):
        if i == 0:
            cur = main_stream
        else:
            cur = side[i - 1]
            cur.wait_stream(main_stream)  # order after the H2D uploads
        n_blocks = n_cells if ids is None else int(ids.numel())
        rc = lib.wva_sweep_launch_bucket(
            ctypes.c_int(n_blocks),
            ctypes.c_int(max(bmax, 1)),
            ctypes.c_int(nt),
            _ptr(ids) if ids is not None else None,
            ctypes.c_int(int(analyzer_mode)),
            ctypes.c_float(float(cv2)),
            ctypes.c_void_p(cur.cuda_stream),
            _ptr(dev["in_tok"]),
            _ptr(dev["out_tok"]),
            _ptr(dev["batch_n"]),
            _ptr(dev["min_replicas"]),
            _ptr(dev["perf_max_batch"]),
            _ptr(dev["cur_replicas"]),
            _ptr(dev["flags"]),
            _ptr(dev["alpha"]),
            _ptr(dev["beta"]),
            _ptr(dev["gamma"]),
            _ptr(dev["delta"]),
            _ptr(dev["arrival_rate"]),
            _ptr(dev["t_itl"]),
            _ptr(dev["t_ttft"]),
            _ptr(dev["t_tps"]),
            _ptr(dev["acc_cost"]),
            _ptr(dev["cur_cost"]),
            _ptr(out.feasible),
            _ptr(out.zero_empty),
            _ptr(out.num_replicas),
            _ptr(out.batch),
            _ptr(out.cost),
            _ptr(out.value),
            _ptr(out.itl),
            _ptr(out.ttft),
            _ptr(out.rho),
            _ptr(out.max_rate),
            _ptr(slabs[0]) if slabs[0] is not None else None,
            _ptr(slabs[1]) if slabs[1] is not None else None,
        )
        if rc != 0:
            raise HipKernelError(f"wva_sweep_launch_bucket(nt={nt}) failed with hipError {rc}")
    for s in side:
        main_stream.wait_stream(s)
    return out


def run_argmin(value, feasible, seg_start) -> "object":
    """Segmented argmin over sweep outputs (device tensors). Returns winner
    cell index (int32, -1 = no feasible candidate) per server, on device."""
    import torch

    lib = load_library()
    n_servers = int(seg_start.shape[0]) - 1
    winner = torch.full((n_servers,), -1, dtype=torch.int32, device=value.device)
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    rc = lib.wva_argmin_launch(
        ctypes.c_int(n_servers),
        stream,
        _ptr(value),
        _ptr(feasible),
        _ptr(seg_start),
        _ptr(winner),
    )
    if rc != 0:
        raise HipKernelError(f"wva_argmin_launch failed with hipError {rc}")
    return winner


def run_greedy_native(cand_value, cand_acc_type, cand_units, cand_replicas,
                      seg_start, srv_priority, capacity, delayed: bool,
                      policy: int, allow_build: bool = True):
    """Host-side native greedy solver (C++ in the same .so; no GPU needed).

    All array args are contiguous numpy arrays (value f32, the rest i32);
    ``capacity`` is mutated in place. Returns (winner_cand, winner_replicas)
    int32 arrays per server (-1 = unallocated). Semantics identical to
    solver.greedy.solve_greedy (differential-tested)."""
    import numpy as np

    lib = load_library(allow_build=allow_build)
    n_servers = len(seg_start) - 1
    out_cand = np.full(n_servers, -1, dtype=np.int32)
    out_reps = np.zeros(n_servers, dtype=np.int32)

    def p(a):
        return a.ctypes.data_as(ctypes.c_void_p)

    rc = lib.wva_greedy_solve(
        ctypes.c_int(n_servers), ctypes.c_int(len(capacity)),
        p(cand_value), p(cand_acc_type), p(cand_units), p(cand_replicas),
        p(seg_start), p(srv_priority), p(capacity),
        ctypes.c_int(1 if delayed else 0), ctypes.c_int(int(policy)),
        p(out_cand), p(out_reps),
    )
    if rc != 0:
        raise HipKernelError(f"wva_greedy_solve failed: {rc}")
    return out_cand, out_reps


def library_loaded() -> bool:
    return _lib is not None and os.path.exists(LIB_PATH)
