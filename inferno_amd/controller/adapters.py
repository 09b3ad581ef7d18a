"""CR/ConfigMap <-> SystemSpec adapters.

Mirrors internal/utils/utils.go:108-383: CreateSystemData (ConfigMaps ->
SystemSpec with Unlimited forced), AddModelAcceleratorProfileToSystemData
(alpha/beta/gamma/delta string parsing), AddServerInfoToSystemData (status
strings -> ServerSpec with KeepAccelerator=true and WVA_SCALE_TO_ZERO),
CreateOptimizedAlloc, FullName, FindModelSLO.
"""
from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass
from functools import lru_cache
from typing import Optional

import yaml

from ..api import v1alpha1 as api
from ..config import (
    AcceleratorSpec,
    AllocationData,
    DecodeParms,
    ModelAcceleratorPerfData,
    ModelTarget,
    OptimizerSpec,
    PrefillParms,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassSpec,
    SystemSpec,
)


class AdapterError(ValueError):
    pass


def full_name(name: str, namespace: str) -> str:
    """Unique server key "name:namespace" (ref utils.go:334)."""
    return f"{name}:{namespace}"


def check_value(x: float) -> bool:
    return not (math.isnan(x) or math.isinf(x))


def _parse_float(s: str, default: float = 0.0) -> float:
    try:
        v = float(s)
    except (TypeError, ValueError):
        return default
    return v if check_value(v) else default


@dataclass
class ServiceClassEntry:
    """yaml shape of one model SLO row (ref internal/interfaces/types.go:20-24)."""

    model: str
    slo_tpot: float = 0.0
    slo_ttft: float = 0.0


def parse_service_class(doc: str) -> tuple[str, int, list[ServiceClassEntry]]:
    """Parse one service-class YAML document.

    Memoized on the document string: the reconciler calls find_model_slo once
    per VA per tick against the same ConfigMap (ref utils.go:369-383 does the
    same re-parse), which made PyYAML the dominant prepare-phase cost at
    fleet scale (measured ~14 ms/VA at 64 VAs). ConfigMap contents change
    rarely, so an LRU on the raw string removes the O(#VAs x #docs) parsing.
    """
    return _parse_service_class_cached(doc)


@lru_cache(maxsize=256)
def _parse_service_class_cached(doc: str) -> tuple[str, int, list[ServiceClassEntry]]:
    sc = yaml.safe_load(doc) or {}
    entries = [
        ServiceClassEntry(
            model=e.get("model", ""),
            slo_tpot=float(e.get("slo-tpot", 0) or 0),
            slo_ttft=float(e.get("slo-ttft", 0) or 0),
        )
        for e in sc.get("data", []) or []
    ]
    # tuple: the memoized value is shared across calls, keep it immutable
    return sc.get("name", ""), int(sc.get("priority", 0) or 0), tuple(entries)


def find_model_slo(cm_data: dict[str, str], target_model: str) -> tuple[ServiceClassEntry, str]:
    """Find the service class entry for a model (ref utils.go:369-383)."""
    for key, val in cm_data.items():
        try:
            class_name, _, entries = parse_service_class(val)
        except yaml.YAMLError as e:
            raise AdapterError(f"failed to parse {key}: {e}")
        for entry in entries:
            if entry.model == target_model:
                return entry, class_name
    raise AdapterError(f"model {target_model!r} not found in any service class")


def create_system_data(
    accelerator_cm: dict[str, str], serviceclass_cm: dict[str, str]
) -> SystemSpec:
    """ConfigMaps -> SystemSpec. Ref utils.go:108-182.

    ``accelerator_cm`` values are JSON blobs {"device": ..., "cost": "40.00"};
    service class values are yaml docs with name/priority/data rows.
    Unlimited mode is forced (utils.go:170-173).
    """
    accelerators = []
    for key, val in accelerator_cm.items():
        try:
            blob = json.loads(val) if isinstance(val, str) else dict(val)
            cost = float(blob["cost"])
        except (json.JSONDecodeError, KeyError, TypeError, ValueError):
            continue  # skip unparseable accelerator (logged by caller)
        accelerators.append(
            AcceleratorSpec(
                name=key,
                type=blob.get("device", ""),
                multiplicity=1,
                cost=cost,
            )
        )

    classes = []
    for key, val in serviceclass_cm.items():
        try:
            name, priority, entries = parse_service_class(val)
        except yaml.YAMLError:
            continue
        classes.append(
            ServiceClassSpec(
                name=name,
                priority=priority,
                modelTargets=[
                    ModelTarget(model=e.model, slo_itl=e.slo_tpot, slo_ttft=e.slo_ttft)
                    for e in entries
                ],
            )
        )

    return SystemSpec(
        accelerators=accelerators,
        models=[],
        serviceClasses=classes,
        servers=[],
        optimizer=OptimizerSpec(unlimited=True),
        capacity=[],
    )


def add_model_accelerator_profile(
    spec: SystemSpec, model_name: str, profile: api.AcceleratorProfile
) -> None:
    """Parse alpha/beta/gamma/delta strings into perf data. Ref utils.go:185-234."""
    decode = profile.perfParms.decodeParms
    if len(decode) < 2:
        raise AdapterError("length of decodeParms should be 2")
    prefill = profile.perfParms.prefillParms
    if len(prefill) < 2:
        raise AdapterError("length of prefillParms should be 2")
    try:
        alpha = float(decode["alpha"])
        beta = float(decode["beta"])
        gamma = float(prefill["gamma"])
        delta = float(prefill["delta"])
    except (KeyError, TypeError, ValueError) as e:
        raise AdapterError(f"invalid perf parms: {e}")
    spec.models.append(
        ModelAcceleratorPerfData(
            name=model_name,
            acc=profile.acc,
            accCount=profile.accCount,
            maxBatchSize=profile.maxBatchSize,
            atTokens=0,
            decodeParms=DecodeParms(alpha=alpha, beta=beta),
            prefillParms=PrefillParms(gamma=gamma, delta=delta),
        )
    )


def add_server_info(
    spec: SystemSpec,
    va: api.VariantAutoscaling,
    class_name: str,
    scale_to_zero: Optional[bool] = None,
) -> None:
    """CR status -> ServerSpec. Ref utils.go:237-311 (KeepAccelerator forced,
    min replicas from WVA_SCALE_TO_ZERO, maxBatchSize from the labeled
    accelerator's profile)."""
    cur = va.status.currentAlloc
    load = ServerLoadSpec(
        arrivalRate=_parse_float(cur.load.arrivalRate),
        avgInTokens=int(_parse_float(cur.load.avgInputTokens)),
        avgOutTokens=int(_parse_float(cur.load.avgOutputTokens)),
    )
    alloc = AllocationData(
        accelerator=cur.accelerator,
        numReplicas=cur.numReplicas,
        maxBatch=cur.maxBatch,
        cost=_parse_float(cur.variantCost),
        itlAverage=_parse_float(cur.itlAverage),
        ttftAverage=_parse_float(cur.ttftAverage),
        load=load,
    )
    if scale_to_zero is None:
        scale_to_zero = os.environ.get("WVA_SCALE_TO_ZERO") == "true"
    server = ServerSpec(
        name=full_name(va.name, va.namespace),
        klass=class_name,
        model=va.spec.modelID,
        keepAccelerator=True,
        minNumReplicas=0 if scale_to_zero else 1,
        currentAlloc=alloc,
        desiredAlloc=AllocationData(),
    )
    acc_name = va.labels.get(api.ACCELERATOR_LABEL, "")
    max_batch = 0
    for ap in va.spec.modelProfile.accelerators:
        if ap.acc == acc_name:
            max_batch = ap.maxBatchSize
            break
    if max_batch > 0:
        server.maxBatchSize = max_batch
    spec.servers.append(server)


def create_optimized_alloc(
    name: str, namespace: str, solution: dict[str, AllocationData], now_iso: str
) -> api.OptimizedAlloc:
    """Solution entry -> OptimizedAlloc. Ref utils.go:314-331."""
    key = full_name(name, namespace)
    if key not in solution:
        raise AdapterError(f"server {key} not found")
    data = solution[key]
    return api.OptimizedAlloc(
        lastRunTime=now_iso,
        accelerator=data.accelerator,
        numReplicas=data.numReplicas,
    )


def get_config_value(data: dict[str, str], key: str, default: str) -> str:
    """Ref utils.go GetConfigValue."""
    return data.get(key, default)


def capacity_from_inventory(
    spec: SystemSpec, inventory: dict[str, dict[str, dict]]
) -> list:
    """Map the cluster GPU inventory (collector.collect_inventory_k8s shape:
    {vendor: {product: {count, memory}}}) onto the SystemSpec's accelerator
    TYPES for the limited-mode solver's capacity constraint.

    A node's ``gpu.product`` matches an accelerator when it equals the
    accelerator's name (the unit-cost ConfigMap key, e.g. "MI355X") or its
    type/device string (e.g. "AMD-MI355X-288GB"). Unmatched products are
    ignored (no accelerator profile to price them)."""
    from ..config import AcceleratorCount

    per_type: dict[str, int] = {}
    for _vendor, products in (inventory or {}).items():
        for product, info in products.items():
            for acc in spec.accelerators:
                if product == acc.name or product == acc.type:
                    per_type[acc.type] = per_type.get(acc.type, 0) + int(
                        info.get("count", 0)
                    )
                    break
    return [AcceleratorCount(type=t, count=c) for t, c in sorted(per_type.items())]
