"""System spec tree — wire-compatible with the reference JSON schema.

Mirrors the reference's ``pkg/config/types.go:6-155`` field-for-field (JSON
key names identical) so that serialized ``SystemData`` documents are
interchangeable between the Go reference and this implementation.
"""
from __future__ import annotations

from dataclasses import dataclass, field, asdict
from typing import Any


@dataclass
class PowerSpec:
    """Accelerator power consumption (Watts). Ref: pkg/config/types.go:47-53."""

    idle: int = 0
    full: int = 0
    midPower: int = 0
    midUtil: float = 0.0


@dataclass
class AcceleratorSpec:
    """One accelerator entry. Ref: pkg/config/types.go:29-45."""

    name: str = ""
    type: str = ""
    multiplicity: int = 1
    memSize: int = 0  # GB
    memBW: int = 0  # GB/s
    power: PowerSpec = field(default_factory=PowerSpec)
    cost: float = 0.0  # cents/hr


@dataclass
class AcceleratorCount:
    """Capacity of an accelerator type. Ref: pkg/config/types.go:60-63."""

    type: str = ""
    count: int = 0


@dataclass
class DecodeParms:
    """decode time = alpha + beta * batch (msec). Ref: pkg/config/types.go:76-80."""

    alpha: float = 0.0
    beta: float = 0.0


@dataclass
class PrefillParms:
    """prefill time = gamma + delta * inTokens * batch (msec). Ref: types.go:82-86."""

    gamma: float = 0.0
    delta: float = 0.0


@dataclass
class ModelAcceleratorPerfData:
    """Perf profile of (model, accelerator). Ref: pkg/config/types.go:64-84."""

    name: str = ""
    acc: str = ""
    accCount: int = 1
    maxBatchSize: int = 0
    atTokens: int = 0
    decodeParms: DecodeParms = field(default_factory=DecodeParms)
    prefillParms: PrefillParms = field(default_factory=PrefillParms)


@dataclass
class ModelTarget:
    """SLO targets of a model within a service class. Ref: types.go:98-104."""

    model: str = ""
    slo_itl: float = 0.0  # json key "slo-itl"
    slo_ttft: float = 0.0  # json key "slo-ttft"
    slo_tps: float = 0.0  # json key "slo-tps"


@dataclass
class ServiceClassSpec:
    """Ref: pkg/config/types.go:92-96."""

    name: str = ""
    priority: int = 0
    modelTargets: list[ModelTarget] = field(default_factory=list)


@dataclass
class ServerLoadSpec:
    """Server load statistics. Ref: pkg/config/types.go:135-139."""

    arrivalRate: float = 0.0  # requests/min
    avgInTokens: int = 0
    avgOutTokens: int = 0


@dataclass
class AllocationData:
    """A server allocation snapshot. Ref: pkg/config/types.go:124-133."""

    accelerator: str = ""
    numReplicas: int = 0
    maxBatch: int = 0
    cost: float = 0.0
    itlAverage: float = 0.0
    ttftAverage: float = 0.0
    load: ServerLoadSpec = field(default_factory=ServerLoadSpec)


@dataclass
class ServerSpec:
    """A deployed variant server. Ref: pkg/config/types.go:112-121."""

    name: str = ""
    klass: str = ""  # json key "class"
    model: str = ""
    keepAccelerator: bool = False
    minNumReplicas: int = 0
    maxBatchSize: int = 0
    currentAlloc: AllocationData = field(default_factory=AllocationData)
    desiredAlloc: AllocationData = field(default_factory=AllocationData)


@dataclass
class OptimizerSpec:
    """Ref: pkg/config/types.go:151-155, extended with the evaluator choice
    (this build): analyzer "mm1k" = the reference's state-dependent M/M/1/K
    chain (default), "mg1" = the closed-form M/G/1/K cheap path with
    squared-CV analyzerCV2 (BASELINE config 4)."""

    unlimited: bool = False
    delayedBestEffort: bool = False
    saturationPolicy: str = "None"
    analyzer: str = "mm1k"
    analyzerCV2: float = 1.0


@dataclass
class SystemSpec:
    """Top-level spec. Ref: pkg/config/types.go:11-22."""

    accelerators: list[AcceleratorSpec] = field(default_factory=list)
    models: list[ModelAcceleratorPerfData] = field(default_factory=list)
    serviceClasses: list[ServiceClassSpec] = field(default_factory=list)
    servers: list[ServerSpec] = field(default_factory=list)
    optimizer: OptimizerSpec = field(default_factory=OptimizerSpec)
    capacity: list[AcceleratorCount] = field(default_factory=list)


# ---------------------------------------------------------------------------
# JSON (de)serialization with the reference's exact key names
# ---------------------------------------------------------------------------

def _mt_to_json(mt: ModelTarget) -> dict[str, Any]:
    return {
        "model": mt.model,
        "slo-itl": mt.slo_itl,
        "slo-ttft": mt.slo_ttft,
        "slo-tps": mt.slo_tps,
    }


def _mt_from_json(d: dict[str, Any]) -> ModelTarget:
    return ModelTarget(
        model=d.get("model", ""),
        slo_itl=float(d.get("slo-itl", 0.0)),
        slo_ttft=float(d.get("slo-ttft", 0.0)),
        slo_tps=float(d.get("slo-tps", 0.0)),
    )


def _server_to_json(s: ServerSpec) -> dict[str, Any]:
    d = asdict(s)
    d["class"] = d.pop("klass")
    return d


def _alloc_from_json(d: dict[str, Any]) -> AllocationData:
    load = d.get("load", {}) or {}
    return AllocationData(
        accelerator=d.get("accelerator", ""),
        numReplicas=int(d.get("numReplicas", 0)),
        maxBatch=int(d.get("maxBatch", 0)),
        cost=float(d.get("cost", 0.0)),
        itlAverage=float(d.get("itlAverage", 0.0)),
        ttftAverage=float(d.get("ttftAverage", 0.0)),
        load=ServerLoadSpec(
            arrivalRate=float(load.get("arrivalRate", 0.0)),
            avgInTokens=int(load.get("avgInTokens", 0)),
            avgOutTokens=int(load.get("avgOutTokens", 0)),
        ),
    )


def system_spec_to_json(spec: SystemSpec) -> dict[str, Any]:
    """Serialize to the reference's ``SystemData`` JSON shape."""
    return {
        "system": {
            "acceleratorData": {"accelerators": [asdict(a) for a in spec.accelerators]},
            "modelData": {"models": [asdict(m) for m in spec.models]},
            "serviceClassData": {
                "serviceClasses": [
                    {
                        "name": sc.name,
                        "priority": sc.priority,
                        "modelTargets": [_mt_to_json(mt) for mt in sc.modelTargets],
                    }
                    for sc in spec.serviceClasses
                ]
            },
            "serverData": {"servers": [_server_to_json(s) for s in spec.servers]},
            "optimizerData": {"optimizer": asdict(spec.optimizer)},
            "capacityData": {"count": [asdict(c) for c in spec.capacity]},
        }
    }


def system_spec_from_json(doc: dict[str, Any]) -> SystemSpec:
    """Parse a reference-format ``SystemData`` JSON document."""
    sys_d = doc.get("system", doc)
    accs = []
    for a in sys_d.get("acceleratorData", {}).get("accelerators", []) or []:
        p = a.get("power", {}) or {}
        accs.append(
            AcceleratorSpec(
                name=a.get("name", ""),
                type=a.get("type", ""),
                multiplicity=int(a.get("multiplicity", 1)),
                memSize=int(a.get("memSize", 0)),
                memBW=int(a.get("memBW", 0)),
                power=PowerSpec(
                    idle=int(p.get("idle", 0)),
                    full=int(p.get("full", 0)),
                    midPower=int(p.get("midPower", 0)),
                    midUtil=float(p.get("midUtil", 0.0)),
                ),
                cost=float(a.get("cost", 0.0)),
            )
        )
    models = []
    for m in sys_d.get("modelData", {}).get("models", []) or []:
        dp = m.get("decodeParms", {}) or {}
        pp = m.get("prefillParms", {}) or {}
        models.append(
            ModelAcceleratorPerfData(
                name=m.get("name", ""),
                acc=m.get("acc", ""),
                accCount=int(m.get("accCount", 1)),
                maxBatchSize=int(m.get("maxBatchSize", 0)),
                atTokens=int(m.get("atTokens", 0)),
                decodeParms=DecodeParms(float(dp.get("alpha", 0.0)), float(dp.get("beta", 0.0))),
                prefillParms=PrefillParms(float(pp.get("gamma", 0.0)), float(pp.get("delta", 0.0))),
            )
        )
    classes = []
    for sc in sys_d.get("serviceClassData", {}).get("serviceClasses", []) or []:
        classes.append(
            ServiceClassSpec(
                name=sc.get("name", ""),
                priority=int(sc.get("priority", 0)),
                modelTargets=[_mt_from_json(mt) for mt in sc.get("modelTargets", []) or []],
            )
        )
    servers = []
    for s in sys_d.get("serverData", {}).get("servers", []) or []:
        servers.append(
            ServerSpec(
                name=s.get("name", ""),
                klass=s.get("class", ""),
                model=s.get("model", ""),
                keepAccelerator=bool(s.get("keepAccelerator", False)),
                minNumReplicas=int(s.get("minNumReplicas", 0)),
                maxBatchSize=int(s.get("maxBatchSize", 0)),
                currentAlloc=_alloc_from_json(s.get("currentAlloc", {}) or {}),
                desiredAlloc=_alloc_from_json(s.get("desiredAlloc", {}) or {}),
            )
        )
    opt_d = sys_d.get("optimizerData", {}).get("optimizer", {}) or {}
    optimizer = OptimizerSpec(
        unlimited=bool(opt_d.get("unlimited", False)),
        delayedBestEffort=bool(opt_d.get("delayedBestEffort", False)),
        saturationPolicy=opt_d.get("saturationPolicy", "None") or "None",
        analyzer=opt_d.get("analyzer", "mm1k") or "mm1k",
        analyzerCV2=float(opt_d.get("analyzerCV2", 1.0)),
    )
    capacity = [
        AcceleratorCount(type=c.get("type", ""), count=int(c.get("count", 0)))
        for c in sys_d.get("capacityData", {}).get("count", []) or []
    ]
    return SystemSpec(
        accelerators=accs,
        models=models,
        serviceClasses=classes,
        servers=servers,
        optimizer=optimizer,
        capacity=capacity,
    )
