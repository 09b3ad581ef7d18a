"""SweepEngine — the batched optimizer engine (GPU or CPU backend).

Replaces the reference's per-server scalar loop
(internal/modelanalyzer/analyzer.go AnalyzeModel -> server.Calculate ->
CreateAllocation per accelerator) with one batched sweep per reconcile:

  * backend "gpu": SoA snapshot upload -> HIP wva_sweep kernel (one workgroup
    per cell) -> wva_argmin segmented winner selection, on the current torch
    HIP stream. Fails loudly if the native library is unavailable.
  * backend "cpu": the golden scalar reference (core.create_allocation per
    cell) — the correctness oracle and the no-GPU fallback
    (condition ``SolverDegraded`` is surfaced by the controller).

Both backends fill ``server.all_allocations`` and (in unlimited mode) select
the min-value candidate per server, so the downstream solver/controller code
is backend-agnostic.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Optional

from ..config import OptimizerSpec, SaturationPolicy
from ..core import Allocation, create_allocation
from ..core.system import System
from ..solver.greedy import solve_greedy
from .snapshot import CellSnapshot, build_cell_snapshot


@dataclass
class EngineStats:
    backend: str = "cpu"
    n_cells: int = 0
    n_servers: int = 0
    sweep_ms: float = 0.0
    solve_ms: float = 0.0
    total_ms: float = 0.0
    extra: dict = field(default_factory=dict)


def _gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


class SweepEngine:
    def __init__(self, backend: str = "auto", device: str = "cuda"):
        if backend == "auto":
            backend = "gpu" if _gpu_available() else "cpu"
        if backend not in ("gpu", "cpu"):
            raise ValueError(f"unknown backend {backend!r}")
        self.backend = backend
        self.device = device
        # set by the reconciler when this engine is a CPU fallback standing
        # in for a failed GPU backend (drives periodic GPU re-probe)
        self.degraded_from_gpu = False

    # ------------------------------------------------------------------
    def sweep(self, system: System, server_names: Optional[list[str]] = None) -> EngineStats:
        """Compute candidate allocations for all (server, accelerator) cells
        and populate ``server.all_allocations``."""
        stats = EngineStats(backend=self.backend)
        t0 = time.perf_counter()
        if self.backend == "gpu":
            snap = build_cell_snapshot(system, server_names)
            stats.n_cells = snap.n_cells
            stats.n_servers = len(snap.server_names)
            t1 = time.perf_counter()
            self._sweep_gpu(system, snap)
            stats.sweep_ms = (time.perf_counter() - t1) * 1000.0
        else:
            names = server_names if server_names is not None else sorted(system.servers)
            stats.n_servers = len(names)
            t1 = time.perf_counter()
            n_cells = 0
            for name in names:
                server = system.servers[name]
                server.all_allocations = {}
                for acc_name in sorted(server.candidate_accelerators(system)):
                    n_cells += 1
                    alloc = create_allocation(system, name, acc_name)
                    if alloc is None:
                        continue
                    if server.cur_allocation is not None:
                        alloc.value = server.cur_allocation.transition_penalty(alloc)
                    server.all_allocations[acc_name] = alloc
            stats.n_cells = n_cells
            stats.sweep_ms = (time.perf_counter() - t1) * 1000.0
        stats.total_ms = (time.perf_counter() - t0) * 1000.0
        return stats

    def _sweep_gpu(self, system: System, snap: CellSnapshot) -> None:
        import torch

        from ..ops.sweep import run_argmin, run_sweep

        if snap.n_cells == 0:
            for name in snap.server_names:
                system.servers[name].all_allocations = {}
            self._last_gpu = None
            return
        mode = 1 if getattr(system, "analyzer_mode", "mm1k") == "mg1" else 0
        out = run_sweep(snap.arrays, device=self.device, analyzer_mode=mode,
                        cv2=getattr(system, "analyzer_cv2", 1.0))
        # keep device-side results for the argmin kernel
        self._last_gpu = (snap, out)
        # download once, build Allocation objects
        cpu = {
            "feasible": out.feasible.cpu().numpy(),
            "zero_empty": out.zero_empty.cpu().numpy(),
            "num_replicas": out.num_replicas.cpu().numpy(),
            "batch": out.batch.cpu().numpy(),
            "cost": out.cost.cpu().numpy(),
            "value": out.value.cpu().numpy(),
            "itl": out.itl.cpu().numpy(),
            "ttft": out.ttft.cpu().numpy(),
            "rho": out.rho.cpu().numpy(),
            "max_rate": out.max_rate.cpu().numpy(),
        }
        torch.cuda.synchronize()
        for name in snap.server_names:
            system.servers[name].all_allocations = {}
        for i in range(snap.n_cells):
            if not cpu["feasible"][i]:
                continue
            server = system.servers[snap.server_names[snap.cell_server[i]]]
            acc_name = "" if cpu["zero_empty"][i] else snap.cell_acc[i]
            alloc = Allocation(
                accelerator=acc_name,
                num_replicas=int(cpu["num_replicas"][i]),
                batch_size=int(cpu["batch"][i]),
                cost=float(cpu["cost"][i]),
                value=float(cpu["value"][i]),
                itl=float(cpu["itl"][i]),
                ttft=float(cpu["ttft"][i]),
                rho=float(cpu["rho"][i]),
                max_arrv_rate_per_replica=float(cpu["max_rate"][i]),
            )
            # keyed by candidate accelerator name (even for the zero-empty
            # allocation, matching server.Calculate's map keying)
            server.all_allocations[snap.cell_acc[i]] = alloc

    # ------------------------------------------------------------------
    def solve(self, system: System, spec: OptimizerSpec) -> EngineStats:
        """Sweep + global solve; sets ``server.allocation`` for every server."""
        stats = self.sweep(system)
        t0 = time.perf_counter()
        if spec.unlimited:
            if self.backend == "gpu" and getattr(self, "_last_gpu", None) is not None:
                self._solve_unlimited_gpu(system)
            else:
                self._solve_unlimited_cpu(system)
        else:
            solve_greedy(
                system,
                delayed_best_effort=spec.delayedBestEffort,
                saturation_policy=SaturationPolicy.parse(spec.saturationPolicy),
            )
        stats.solve_ms = (time.perf_counter() - t0) * 1000.0
        stats.total_ms += stats.solve_ms
        system.allocate_by_type()
        return stats

    def _solve_unlimited_cpu(self, system: System) -> None:
        for server in system.servers.values():
            server.remove_allocation()
            best = None
            for acc_name in sorted(server.all_allocations):
                alloc = server.all_allocations[acc_name]
                if best is None or alloc.value < best.value:
                    best = alloc
            if best is not None:
                server.set_allocation(best)

    def _solve_unlimited_gpu(self, system: System) -> None:
        from ..ops.sweep import run_argmin

        snap, out = self._last_gpu
        seg = snap.seg_start.to(out.value.device)
        winner = run_argmin(out.value, out.feasible, seg).cpu().numpy()
        for s, name in enumerate(snap.server_names):
            server = system.servers[name]
            server.remove_allocation()
            w = int(winner[s])
            if w < 0:
                continue
            acc_key = snap.cell_acc[w]
            alloc = server.all_allocations.get(acc_key)
            if alloc is not None:
                server.set_allocation(alloc)
