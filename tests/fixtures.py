"""Shared system fixtures for core/solver/engine tests."""
from __future__ import annotations

import numpy as np

from inferno_amd.config import (
    AcceleratorCount,
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

ACCELERATORS = [
    # name, type, cost (cents/hr)
    ("MI355X", "AMD-MI355X-288GB", 90.0),
    ("MI325X", "AMD-MI325X-256GB", 70.0),
    ("MI300X", "AMD-MI300X-192GB", 65.0),
]


def make_spec(
    n_servers: int = 4,
    seed: int = 0,
    unlimited: bool = True,
    saturation_policy: str = "None",
    delayed_best_effort: bool = False,
    capacity: dict | None = None,
    arrival_scale: float = 60.0,
    min_num_replicas: int = 1,
    keep_accelerator: bool = False,
    n_accelerators: int = 3,
    priorities=(1, 10),
) -> SystemSpec:
    """Build a randomized synthetic fleet spec shaped like the reference's configs."""
    rng = np.random.default_rng(seed)
    accs = [
        AcceleratorSpec(name=n, type=t, multiplicity=1, cost=c)
        for n, t, c in ACCELERATORS[:n_accelerators]
    ]
    classes = []
    models = []
    servers = []
    class_names = []
    for ci, prio in enumerate(priorities):
        class_names.append(f"class-{ci}")
        classes.append(ServiceClassSpec(name=f"class-{ci}", priority=prio, modelTargets=[]))

    for i in range(n_servers):
        model_name = f"model-{i}"
        ci = i % len(priorities)
        itl = float(rng.uniform(25, 200))
        ttft = float(rng.uniform(500, 3000))
        classes[ci].modelTargets.append(
            ModelTarget(model=model_name, slo_itl=itl, slo_ttft=ttft, slo_tps=0.0)
        )
        for a in accs:
            scale = 1.0 + 0.3 * (accs.index(a))
            models.append(
                ModelAcceleratorPerfData(
                    name=model_name,
                    acc=a.name,
                    accCount=int(rng.choice([1, 1, 2, 4])),
                    maxBatchSize=int(rng.choice([64, 128, 256])),
                    atTokens=int(rng.choice([512, 1024, 2048])),
                    decodeParms=DecodeParms(
                        alpha=float(rng.uniform(5, 25)) * scale,
                        beta=float(rng.uniform(0.05, 0.5)) * scale,
                    ),
                    prefillParms=PrefillParms(
                        gamma=float(rng.uniform(1, 10)) * scale,
                        delta=float(rng.uniform(1e-4, 1e-2)) * scale,
                    ),
                )
            )
        cur_acc = accs[i % len(accs)].name
        servers.append(
            ServerSpec(
                name=f"srv-{i}:ns",
                klass=class_names[ci],
                model=model_name,
                keepAccelerator=keep_accelerator,
                minNumReplicas=min_num_replicas,
                maxBatchSize=0,
                currentAlloc=AllocationData(
                    accelerator=cur_acc,
                    numReplicas=1,
                    maxBatch=256,
                    cost=float(accs[i % len(accs)].cost),
                    load=ServerLoadSpec(
                        arrivalRate=float(rng.uniform(0.5, 4.0)) * arrival_scale,
                        avgInTokens=int(rng.integers(64, 2048)),
                        avgOutTokens=int(rng.integers(32, 512)),
                    ),
                ),
            )
        )
    cap = capacity or {}
    return SystemSpec(
        accelerators=accs,
        models=models,
        serviceClasses=classes,
        servers=servers,
        optimizer=OptimizerSpec(
            unlimited=unlimited,
            delayedBestEffort=delayed_best_effort,
            saturationPolicy=saturation_policy,
        ),
        capacity=[AcceleratorCount(type=t, count=c) for t, c in cap.items()],
    )
