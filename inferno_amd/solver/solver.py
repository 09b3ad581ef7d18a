"""Global allocation solvers.

``Solver`` mirrors the reference's pkg/solver/solver.go (snapshot + unlimited
per-server argmin + diff map). The greedy capacity-constrained path lives in
``greedy.py``. On the GPU path the unlimited argmin is replaced by HIP kernel
K3 (segmented argmin over the sweep output); this module is the semantic
oracle and the CPU fallback.

Tie-breaking note: the reference iterates Go maps (randomized order) with a
strict ``<`` argmin (solver.go:63-79), so ties are nondeterministic there.
Here candidates are visited in sorted accelerator-name order, making ties
deterministic (first name wins) — any reference tie outcome is equally valid.
"""
from __future__ import annotations

import time
from typing import Optional

from ..config import OptimizerSpec, SaturationPolicy
from ..core import Allocation, AllocationDiff, create_allocation_diff
from ..core.system import System
from .greedy import solve_greedy

_MAX_FLOAT32 = 3.4028234663852886e38


class Solver:
    """Ref: pkg/solver/solver.go:13-58."""

    def __init__(self, optimizer_spec: OptimizerSpec):
        self.optimizer_spec = optimizer_spec
        self.current_allocation: dict[str, Allocation] = {}
        self.diff_allocation: dict[str, AllocationDiff] = {}

    def solve(self, system: System) -> None:
        # snapshot of current allocations
        self.current_allocation = {}
        for name, server in system.servers.items():
            if server.cur_allocation is not None:
                self.current_allocation[name] = server.cur_allocation

        if self.optimizer_spec.unlimited:
            self.solve_unlimited(system)
        else:
            solve_greedy(
                system,
                delayed_best_effort=self.optimizer_spec.delayedBestEffort,
                saturation_policy=SaturationPolicy.parse(self.optimizer_spec.saturationPolicy),
            )

        self.diff_allocation = {}
        for name, server in system.servers.items():
            cur = self.current_allocation.get(name)
            desired = server.allocation
            diff = create_allocation_diff(cur, desired)
            if diff is not None:
                self.diff_allocation[name] = diff

    def solve_unlimited(self, system: System) -> None:
        """Per-server argmin over candidate allocations by value. Ref: solver.go:63-79."""
        for server in system.servers.values():
            server.remove_allocation()
            min_val = _MAX_FLOAT32
            min_alloc: Optional[Allocation] = None
            for acc_name in sorted(server.all_allocations):
                alloc = server.all_allocations[acc_name]
                if alloc.value < min_val:
                    min_val = alloc.value
                    min_alloc = alloc
            if min_alloc is not None:
                server.set_allocation(min_alloc)


class Optimizer:
    """Timed solve wrapper. Ref: pkg/solver/optimizer.go:11-49."""

    def __init__(self, spec: OptimizerSpec):
        self.spec = spec
        self.solver: Optional[Solver] = None
        self.solution_time_msec: float = 0.0

    def optimize(self, system: System) -> None:
        if self.spec is None:
            raise ValueError("missing optimizer spec")
        self.solver = Solver(self.spec)
        t0 = time.perf_counter()
        self.solver.solve(system)
        self.solution_time_msec = (time.perf_counter() - t0) * 1000.0


class Manager:
    """Glue: optimize + per-type aggregation. Ref: pkg/manager/manager.go (sans singleton)."""

    def __init__(self, system: System, optimizer: Optimizer):
        self.system = system
        self.optimizer = optimizer

    def optimize(self) -> None:
        self.optimizer.optimize(self.system)
        self.system.allocate_by_type()
