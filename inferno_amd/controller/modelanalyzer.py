"""Model analyzer + optimizer engine adapters.

Parity layer for the reference's internal/modelanalyzer/analyzer.go (thin
wrapper exposing per-VA candidate allocations as a ModelAnalyzeResponse) and
internal/optimizer/optimizer.go (VariantAutoscalingsEngine.Optimize mapping
the global solution to per-VA OptimizedAllocs). The reconciler's batched
path (ShardedSolver) supersedes these in the hot loop; they are the
per-variant inspection API (used by tests/tooling and kept for interface
compatibility).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Optional

from ..api import v1alpha1 as api
from ..core import Allocation
from ..core.system import System
from ..engine import SweepEngine
from ..solver import Manager
from .adapters import full_name


@dataclass
class ModelAcceleratorAllocation:
    """Ref: internal/interfaces/types.go:12-18."""

    allocation: Allocation
    reason: str = "markovian analysis"


@dataclass
class ModelAnalyzeResponse:
    """Feasible allocations per accelerator. Ref: interfaces/types.go:6-9."""

    allocations: dict[str, ModelAcceleratorAllocation] = field(default_factory=dict)


class ModelAnalyzer:
    """Per-variant queueing analysis. Ref: internal/modelanalyzer/analyzer.go:14-35."""

    def __init__(self, system: System, engine: Optional[SweepEngine] = None):
        self.system = system
        self.engine = engine or SweepEngine(backend="cpu")

    def analyze_model(self, va: api.VariantAutoscaling) -> ModelAnalyzeResponse:
        server_name = full_name(va.name, va.namespace)
        server = self.system.servers.get(server_name)
        if server is None:
            return ModelAnalyzeResponse()
        self.engine.sweep(self.system, server_names=[server_name])
        return ModelAnalyzeResponse(
            allocations={
                acc: ModelAcceleratorAllocation(allocation=alloc)
                for acc, alloc in server.all_allocations.items()
            }
        )


class VariantAutoscalingsEngine:
    """Global optimization -> per-VA OptimizedAlloc map.

    Ref: internal/optimizer/optimizer.go:17-54 (keyed by VA NAME, matching
    applyOptimizedAllocations' lookup).
    """

    def __init__(self, manager: Manager, system: System):
        self.manager = manager
        self.system = system

    def optimize(
        self, va_list: list[api.VariantAutoscaling],
        analysis: Optional[dict[str, ModelAnalyzeResponse]] = None,
    ) -> dict[str, api.OptimizedAlloc]:
        self.manager.optimize()
        solution = self.system.generate_solution()
        if not solution:
            raise RuntimeError("no feasible allocations found for all variants")
        now = datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")
        out: dict[str, api.OptimizedAlloc] = {}
        for va in va_list:
            key = full_name(va.name, va.namespace)
            data = solution.get(key)
            if data is None:
                continue
            out[va.name] = api.OptimizedAlloc(
                lastRunTime=now,
                accelerator=data.accelerator,
                numReplicas=data.numReplicas,
            )
        return out
