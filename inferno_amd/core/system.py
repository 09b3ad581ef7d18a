"""Domain model: System / Accelerator / Model / ServiceClass / Server.

Re-design of the reference's ``pkg/core`` WITHOUT the package-global
singleton (``core.TheSystem``, system.go:10-45) — every operation takes the
``System`` explicitly, so many system snapshots can be analyzed concurrently
(required by the batched GPU sweep and by multi-GPU sharding).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from ..config import (
    DEFAULT_HIGH_PRIORITY,
    DEFAULT_LOW_PRIORITY,
    DEFAULT_SERVICE_CLASS_NAME,
    DEFAULT_SERVICE_CLASS_PRIORITY,
    AcceleratorSpec,
    AllocationData,
    ModelAcceleratorPerfData,
    OptimizerSpec,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassSpec,
    SystemSpec,
)
from .allocation import Allocation, allocation_from_data, create_allocation


class Accelerator:
    """Ref: pkg/core/accelerator.go:12-71 (piecewise-linear power model)."""

    def __init__(self, spec: AcceleratorSpec):
        self.name = spec.name
        self.spec = spec
        self._slope_low = 0.0
        self._slope_high = 0.0

    def calculate(self) -> None:
        p = self.spec.power
        if p.midUtil > 0:
            self._slope_low = (p.midPower - p.idle) / p.midUtil
        if p.midUtil < 1:
            self._slope_high = (p.full - p.midPower) / (1.0 - p.midUtil)

    def power(self, util: float) -> float:
        p = self.spec.power
        if util <= p.midUtil:
            return p.idle + self._slope_low * util
        return p.midPower + self._slope_high * (util - p.midUtil)

    @property
    def type(self) -> str:
        return self.spec.type

    @property
    def cost(self) -> float:
        return self.spec.cost

    @property
    def multiplicity(self) -> int:
        return self.spec.multiplicity

    @property
    def mem_size(self) -> int:
        return self.spec.memSize


class Model:
    """Per-accelerator perf data + instances needed. Ref: pkg/core/model.go."""

    def __init__(self, name: str):
        self.name = name
        self.perf_data: dict[str, ModelAcceleratorPerfData] = {}
        self.num_instances: dict[str, int] = {}

    def add_perf_data(self, spec: ModelAcceleratorPerfData) -> None:
        if spec.name != self.name:
            return
        self.perf_data[spec.acc] = spec
        count = spec.accCount if spec.accCount > 0 else 1
        self.num_instances[spec.acc] = count

    def get_perf_data(self, acc_name: str) -> Optional[ModelAcceleratorPerfData]:
        return self.perf_data.get(acc_name)

    def get_num_instances(self, acc_name: str) -> int:
        return self.num_instances.get(acc_name, 0)


@dataclass
class Target:
    """SLO targets for a (service class, model). Ref: pkg/core/serviceclass.go:16-20."""

    itl: float = 0.0
    ttft: float = 0.0
    tps: float = 0.0


class ServiceClass:
    """Ref: pkg/core/serviceclass.go."""

    def __init__(self, name: str, priority: int):
        if priority < DEFAULT_HIGH_PRIORITY or priority > DEFAULT_LOW_PRIORITY:
            priority = DEFAULT_SERVICE_CLASS_PRIORITY
        self.name = name
        self.priority = priority
        self.targets: dict[str, Target] = {}

    @classmethod
    def from_spec(cls, spec: ServiceClassSpec) -> "ServiceClass":
        svc = cls(spec.name, spec.priority)
        for mt in spec.modelTargets:
            svc.targets[mt.model] = Target(itl=mt.slo_itl, ttft=mt.slo_ttft, tps=mt.slo_tps)
        return svc

    def model_target(self, model_name: str) -> Optional[Target]:
        return self.targets.get(model_name)


class Server:
    """A deployed variant (service class + model + load). Ref: pkg/core/server.go."""

    def __init__(self, spec: ServerSpec):
        self.name = spec.name
        self.service_class_name = spec.klass or DEFAULT_SERVICE_CLASS_NAME
        self.model_name = spec.model
        self.keep_accelerator = spec.keepAccelerator
        self.min_num_replicas = spec.minNumReplicas
        self.max_batch_size = spec.maxBatchSize
        self.load: ServerLoadSpec = spec.currentAlloc.load
        self.all_allocations: dict[str, Allocation] = {}
        self.allocation: Optional[Allocation] = None
        self.cur_allocation: Optional[Allocation] = allocation_from_data(spec.currentAlloc)
        self.spec = spec

    def priority(self, system: "System") -> int:
        svc = system.service_classes.get(self.service_class_name)
        return svc.priority if svc is not None else DEFAULT_SERVICE_CLASS_PRIORITY

    def candidate_accelerators(self, system: "System") -> dict[str, Accelerator]:
        """Ref: server.go:70-82 — KeepAccelerator restricts to current acc."""
        accs = system.accelerators
        if self.keep_accelerator and self.cur_allocation is not None and self.cur_allocation.accelerator:
            cur = accs.get(self.cur_allocation.accelerator)
            return {cur.name: cur} if cur is not None else {}
        return accs

    def calculate(self, system: "System") -> None:
        """Build one candidate allocation per candidate accelerator.

        Value = transition penalty from current allocation (server.go:55-67).
        """
        self.all_allocations = {}
        for g in self.candidate_accelerators(system).values():
            alloc = create_allocation(system, self.name, g.name)
            if alloc is None:
                continue
            if self.cur_allocation is not None:
                alloc.value = self.cur_allocation.transition_penalty(alloc)
            self.all_allocations[g.name] = alloc

    def set_allocation(self, alloc: Optional[Allocation]) -> None:
        self.allocation = alloc
        self.update_desired_alloc()

    def remove_allocation(self) -> None:
        self.allocation = None

    def saturated(self) -> bool:
        return (
            self.allocation is not None
            and self.load is not None
            and self.allocation.is_saturated(self.load.arrivalRate)
        )

    def update_desired_alloc(self) -> None:
        if self.allocation is not None:
            self.spec.desiredAlloc = self.allocation.to_data()
            self.spec.desiredAlloc.load = self.load
        else:
            self.spec.desiredAlloc = AllocationData()

    def apply_desired_alloc(self) -> None:
        self.spec.currentAlloc = self.spec.desiredAlloc
        self.cur_allocation = allocation_from_data(self.spec.currentAlloc)
        self.load = self.spec.currentAlloc.load


@dataclass
class AllocationByType:
    """Aggregated allocation per accelerator type. Ref: system.go:58-66."""

    name: str
    count: int = 0
    limit: int = 0
    cost: float = 0.0


class System:
    """Registry of accelerators/models/classes/servers. Reentrant (no singleton)."""

    def __init__(self) -> None:
        self.analyzer_mode: str = "mm1k"  # or "mg1" (closed-form cheap path)
        self.analyzer_cv2: float = 1.0
        self.accelerators: dict[str, Accelerator] = {}
        self.models: dict[str, Model] = {}
        self.service_classes: dict[str, ServiceClass] = {}
        self.servers: dict[str, Server] = {}
        self.capacity: dict[str, int] = {}
        self.allocation_by_type: dict[str, AllocationByType] = {}

    @classmethod
    def from_spec(cls, spec: SystemSpec) -> tuple["System", OptimizerSpec]:
        """Ref: system.go:82-92 SetFromSpec."""
        system = cls()
        for a in spec.accelerators:
            system.accelerators[a.name] = Accelerator(a)
        for pd in spec.models:
            model = system.models.get(pd.name)
            if model is None:
                model = Model(pd.name)
                system.models[pd.name] = model
            model.add_perf_data(pd)
        for sc in spec.serviceClasses:
            system.service_classes[sc.name] = ServiceClass.from_spec(sc)
        for sv in spec.servers:
            system.servers[sv.name] = Server(sv)
        for c in spec.capacity:
            system.capacity[c.type] = c.count
        system.analyzer_mode = spec.optimizer.analyzer or "mm1k"
        system.analyzer_cv2 = spec.optimizer.analyzerCV2
        return system, spec.optimizer

    def calculate(self) -> None:
        """Compute candidate allocations for every server. Ref: system.go:259-269."""
        for g in self.accelerators.values():
            g.calculate()
        for v in self.servers.values():
            v.calculate(self)

    def allocate_by_type(self) -> dict[str, AllocationByType]:
        """Accumulate solution allocations per accelerator type. Ref: system.go:271-300."""
        self.allocation_by_type = {}
        for server in self.servers.values():
            alloc = server.allocation
            if alloc is None:
                continue
            acc = self.accelerators.get(alloc.accelerator)
            model = self.models.get(server.model_name)
            if acc is None or model is None:
                continue
            t = acc.type
            agg = self.allocation_by_type.get(t)
            if agg is None:
                agg = AllocationByType(name=t, limit=self.capacity.get(t, 0))
            agg.count += alloc.num_replicas * model.get_num_instances(acc.name) * acc.multiplicity
            agg.cost += alloc.cost
            self.allocation_by_type[t] = agg
        return self.allocation_by_type

    def generate_solution(self) -> dict[str, AllocationData]:
        """JSON allocation map for all servers with a solution. Ref: system.go:303-319."""
        solution: dict[str, AllocationData] = {}
        for name, server in self.servers.items():
            alloc = server.allocation
            if alloc is None:
                continue
            data = alloc.to_data()
            data.load = server.load
            solution[name] = data
        return solution
