"""llmd.ai/v1alpha1 VariantAutoscaling API types — wire-identical to the
reference CRD (api/v1alpha1/variantautoscaling_types.go:8-222): same group,
version, kind, shortName, JSON field names, status string formats and
condition types/reasons, so existing CRs, HPA/KEDA pipelines and kubectl
printer columns drop in unchanged.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional

GROUP = "llmd.ai"
VERSION = "v1alpha1"
KIND = "VariantAutoscaling"
PLURAL = "variantautoscalings"
SHORT_NAME = "va"
API_VERSION = f"{GROUP}/{VERSION}"

# the label naming the accelerator used by a variant
# (ref internal/collector/collector.go:250 & utils.go:296)
ACCELERATOR_LABEL = "inference.optimization/acceleratorName"

# Condition types (ref variantautoscaling_types.go:195-201)
TYPE_METRICS_AVAILABLE = "MetricsAvailable"
TYPE_OPTIMIZATION_READY = "OptimizationReady"

# Condition reasons (ref variantautoscaling_types.go:203-222)
REASON_METRICS_FOUND = "MetricsFound"
REASON_METRICS_MISSING = "MetricsMissing"
REASON_METRICS_STALE = "MetricsStale"
REASON_PROMETHEUS_ERROR = "PrometheusError"
REASON_OPTIMIZATION_SUCCEEDED = "OptimizationSucceeded"
REASON_OPTIMIZATION_FAILED = "OptimizationFailed"
REASON_METRICS_UNAVAILABLE = "MetricsUnavailable"
# this build only: GPU solver fell back to the CPU reference path
REASON_SOLVER_DEGRADED = "SolverDegraded"


def _now_iso() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


@dataclass
class ConfigMapKeyRef:
    name: str = ""
    key: str = ""


@dataclass
class PerfParms:
    """decodeParms: alpha/beta strings; prefillParms: gamma/delta strings
    (string-typed maps on the wire, ref types.go:41-50)."""

    decodeParms: dict[str, str] = field(default_factory=dict)
    prefillParms: dict[str, str] = field(default_factory=dict)


@dataclass
class AcceleratorProfile:
    acc: str = ""
    accCount: int = 1
    perfParms: PerfParms = field(default_factory=PerfParms)
    maxBatchSize: int = 1


@dataclass
class ModelProfile:
    accelerators: list[AcceleratorProfile] = field(default_factory=list)


@dataclass
class VariantAutoscalingSpec:
    modelID: str = ""
    sloClassRef: ConfigMapKeyRef = field(default_factory=ConfigMapKeyRef)
    modelProfile: ModelProfile = field(default_factory=ModelProfile)


@dataclass
class LoadProfile:
    """String-typed load stats (ref types.go:126-135)."""

    arrivalRate: str = ""
    avgInputTokens: str = ""
    avgOutputTokens: str = ""


@dataclass
class Allocation:
    """Current allocation status; floats carried as 2-decimal strings
    (ref collector.go:268-275)."""

    accelerator: str = ""
    numReplicas: int = 0
    maxBatch: int = 0
    variantCost: str = "0.00"
    itlAverage: str = "0.00"
    ttftAverage: str = "0.00"
    load: LoadProfile = field(default_factory=LoadProfile)


@dataclass
class OptimizedAlloc:
    lastRunTime: str = ""
    accelerator: str = ""
    numReplicas: int = 0


@dataclass
class ActuationStatus:
    applied: bool = False


@dataclass
class Condition:
    type: str = ""
    status: str = "Unknown"  # "True" | "False" | "Unknown"
    observedGeneration: int = 0
    lastTransitionTime: str = ""
    reason: str = ""
    message: str = ""


@dataclass
class VariantAutoscalingStatus:
    currentAlloc: Allocation = field(default_factory=Allocation)
    desiredOptimizedAlloc: OptimizedAlloc = field(default_factory=OptimizedAlloc)
    actuation: ActuationStatus = field(default_factory=ActuationStatus)
    conditions: list[Condition] = field(default_factory=list)


@dataclass
class VariantAutoscaling:
    name: str = ""
    namespace: str = "default"
    labels: dict[str, str] = field(default_factory=dict)
    generation: int = 1
    uid: str = ""
    resourceVersion: str = ""
    ownerReferences: list[dict] = field(default_factory=list)
    deletionTimestamp: Optional[str] = None
    spec: VariantAutoscalingSpec = field(default_factory=VariantAutoscalingSpec)
    status: VariantAutoscalingStatus = field(default_factory=VariantAutoscalingStatus)

    @property
    def accelerator_name(self) -> str:
        return self.labels.get(ACCELERATOR_LABEL, "")


# ---------------------------------------------------------------------------
# condition helpers (ref api/v1alpha1/conditions.go:9-34 + apimachinery meta)
# ---------------------------------------------------------------------------

def set_condition(va: VariantAutoscaling, ctype: str, status: str, reason: str,
                  message: str) -> None:
    """Set/update a condition; lastTransitionTime only changes when status
    flips (meta.SetStatusCondition semantics)."""
    for c in va.status.conditions:
        if c.type == ctype:
            if c.status != status:
                c.lastTransitionTime = _now_iso()
            c.status = status
            c.reason = reason
            c.message = message
            c.observedGeneration = va.generation
            return
    va.status.conditions.append(
        Condition(
            type=ctype,
            status=status,
            observedGeneration=va.generation,
            lastTransitionTime=_now_iso(),
            reason=reason,
            message=message,
        )
    )


def get_condition(va: VariantAutoscaling, ctype: str) -> Optional[Condition]:
    for c in va.status.conditions:
        if c.type == ctype:
            return c
    return None


def is_condition_true(va: VariantAutoscaling, ctype: str) -> bool:
    c = get_condition(va, ctype)
    return c is not None and c.status == "True"


def is_condition_false(va: VariantAutoscaling, ctype: str) -> bool:
    c = get_condition(va, ctype)
    return c is not None and c.status == "False"


# ---------------------------------------------------------------------------
# JSON (de)serialization — k8s wire shape
# ---------------------------------------------------------------------------

def va_to_json(va: VariantAutoscaling) -> dict[str, Any]:
    meta: dict[str, Any] = {
        "name": va.name,
        "namespace": va.namespace,
        "labels": dict(va.labels),
        "generation": va.generation,
    }
    if va.uid:
        meta["uid"] = va.uid
    if va.resourceVersion:
        meta["resourceVersion"] = va.resourceVersion
    if va.ownerReferences:
        meta["ownerReferences"] = va.ownerReferences
    if va.deletionTimestamp:
        meta["deletionTimestamp"] = va.deletionTimestamp
    return {
        "apiVersion": API_VERSION,
        "kind": KIND,
        "metadata": meta,
        "spec": {
            "modelID": va.spec.modelID,
            "sloClassRef": {
                "name": va.spec.sloClassRef.name,
                "key": va.spec.sloClassRef.key,
            },
            "modelProfile": {
                "accelerators": [
                    {
                        "acc": ap.acc,
                        "accCount": ap.accCount,
                        "perfParms": {
                            "decodeParms": dict(ap.perfParms.decodeParms),
                            "prefillParms": dict(ap.perfParms.prefillParms),
                        },
                        "maxBatchSize": ap.maxBatchSize,
                    }
                    for ap in va.spec.modelProfile.accelerators
                ]
            },
        },
        "status": {
            "currentAlloc": {
                "accelerator": va.status.currentAlloc.accelerator,
                "numReplicas": va.status.currentAlloc.numReplicas,
                "maxBatch": va.status.currentAlloc.maxBatch,
                "variantCost": va.status.currentAlloc.variantCost,
                "itlAverage": va.status.currentAlloc.itlAverage,
                "ttftAverage": va.status.currentAlloc.ttftAverage,
                "load": {
                    "arrivalRate": va.status.currentAlloc.load.arrivalRate,
                    "avgInputTokens": va.status.currentAlloc.load.avgInputTokens,
                    "avgOutputTokens": va.status.currentAlloc.load.avgOutputTokens,
                },
            },
            "desiredOptimizedAlloc": {
                "lastRunTime": va.status.desiredOptimizedAlloc.lastRunTime,
                "accelerator": va.status.desiredOptimizedAlloc.accelerator,
                "numReplicas": va.status.desiredOptimizedAlloc.numReplicas,
            },
            "actuation": {"applied": va.status.actuation.applied},
            "conditions": [
                {
                    "type": c.type,
                    "status": c.status,
                    "observedGeneration": c.observedGeneration,
                    "lastTransitionTime": c.lastTransitionTime,
                    "reason": c.reason,
                    "message": c.message,
                }
                for c in va.status.conditions
            ],
        },
    }


def va_from_json(doc: dict[str, Any]) -> VariantAutoscaling:
    meta = doc.get("metadata", {}) or {}
    spec_d = doc.get("spec", {}) or {}
    status_d = doc.get("status", {}) or {}
    slo = spec_d.get("sloClassRef", {}) or {}
    profile = spec_d.get("modelProfile", {}) or {}
    accs = []
    for ap in profile.get("accelerators", []) or []:
        pp = ap.get("perfParms", {}) or {}
        accs.append(
            AcceleratorProfile(
                acc=ap.get("acc", ""),
                accCount=int(ap.get("accCount", 1)),
                perfParms=PerfParms(
                    decodeParms=dict(pp.get("decodeParms", {}) or {}),
                    prefillParms=dict(pp.get("prefillParms", {}) or {}),
                ),
                maxBatchSize=int(ap.get("maxBatchSize", 1)),
            )
        )
    cur_d = status_d.get("currentAlloc", {}) or {}
    load_d = cur_d.get("load", {}) or {}
    des_d = status_d.get("desiredOptimizedAlloc", {}) or {}
    conds = [
        Condition(
            type=c.get("type", ""),
            status=c.get("status", "Unknown"),
            observedGeneration=int(c.get("observedGeneration", 0)),
            lastTransitionTime=c.get("lastTransitionTime", ""),
            reason=c.get("reason", ""),
            message=c.get("message", ""),
        )
        for c in status_d.get("conditions", []) or []
    ]
    return VariantAutoscaling(
        name=meta.get("name", ""),
        namespace=meta.get("namespace", "default"),
        labels=dict(meta.get("labels", {}) or {}),
        generation=int(meta.get("generation", 1)),
        uid=meta.get("uid", "") or "",
        resourceVersion=str(meta.get("resourceVersion", "") or ""),
        ownerReferences=list(meta.get("ownerReferences", []) or []),
        deletionTimestamp=meta.get("deletionTimestamp"),
        spec=VariantAutoscalingSpec(
            modelID=spec_d.get("modelID", ""),
            sloClassRef=ConfigMapKeyRef(name=slo.get("name", ""), key=slo.get("key", "")),
            modelProfile=ModelProfile(accelerators=accs),
        ),
        status=VariantAutoscalingStatus(
            currentAlloc=Allocation(
                accelerator=cur_d.get("accelerator", ""),
                numReplicas=int(cur_d.get("numReplicas", 0)),
                maxBatch=int(cur_d.get("maxBatch", 0)),
                variantCost=cur_d.get("variantCost", "0.00"),
                itlAverage=cur_d.get("itlAverage", "0.00"),
                ttftAverage=cur_d.get("ttftAverage", "0.00"),
                load=LoadProfile(
                    arrivalRate=load_d.get("arrivalRate", ""),
                    avgInputTokens=load_d.get("avgInputTokens", ""),
                    avgOutputTokens=load_d.get("avgOutputTokens", ""),
                ),
            ),
            desiredOptimizedAlloc=OptimizedAlloc(
                lastRunTime=des_d.get("lastRunTime", ""),
                accelerator=des_d.get("accelerator", ""),
                numReplicas=int(des_d.get("numReplicas", 0)),
            ),
            actuation=ActuationStatus(
                applied=bool((status_d.get("actuation", {}) or {}).get("applied", False))
            ),
            conditions=conds,
        ),
    )
