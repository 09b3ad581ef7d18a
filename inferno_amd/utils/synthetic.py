"""Synthetic fleet + trace generation for benchmarks and tests.

Generates SystemSpec fleets shaped like the reference's sweep configs
(BASELINE.json): N models x {MI300X, MI325X, MI355X} accelerator variants,
Premium/Freemium service classes, and bursty Poisson request traces with
piecewise rate schedules (semantics of tools/vllm-emulator/loadgen.py:10-18).
"""
from __future__ import annotations

import numpy as np

from ..config import (
    AcceleratorCount,
    AcceleratorSpec,
    AllocationData,
    DecodeParms,
    ModelAcceleratorPerfData,
    ModelTarget,
    OptimizerSpec,
    PowerSpec,
    PrefillParms,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassSpec,
    SystemSpec,
)

# AMD Instinct accelerator table (costs follow the reference ConfigMap's
# MI300X=65c/hr anchor, deploy/configmap-accelerator-unitcost.yaml:15-28)
AMD_ACCELERATORS = [
    AcceleratorSpec(
        name="MI300X",
        type="AMD-MI300X-192GB",
        multiplicity=1,
        memSize=192,
        memBW=5300,
        power=PowerSpec(idle=140, full=750, midPower=550, midUtil=0.6),
        cost=65.0,
    ),
    AcceleratorSpec(
        name="MI325X",
        type="AMD-MI325X-256GB",
        multiplicity=1,
        memSize=256,
        memBW=6000,
        power=PowerSpec(idle=150, full=1000, midPower=700, midUtil=0.6),
        cost=78.0,
    ),
    AcceleratorSpec(
        name="MI355X",
        type="AMD-MI355X-288GB",
        multiplicity=1,
        memSize=288,
        memBW=8000,
        power=PowerSpec(idle=180, full=1400, midPower=950, midUtil=0.55),
        cost=95.0,
    ),
]


def make_fleet_spec(
    n_models: int,
    seed: int = 0,
    accelerators: list[AcceleratorSpec] | None = None,
    min_num_replicas: int = 1,
    unlimited: bool = True,
    capacity: dict[str, int] | None = None,
    perf_profiles: dict[str, list[ModelAcceleratorPerfData]] | None = None,
) -> SystemSpec:
    """Random-init fleet of ``n_models`` variants over the AMD accelerator set.

    ``perf_profiles`` optionally supplies derived (e.g. MI355X roofline-based)
    perf entries per model instead of random ones.
    """
    rng = np.random.default_rng(seed)
    accs = accelerators if accelerators is not None else AMD_ACCELERATORS
    classes = [
        ServiceClassSpec(name="Premium", priority=1, modelTargets=[]),
        ServiceClassSpec(name="Freemium", priority=10, modelTargets=[]),
    ]
    models: list[ModelAcceleratorPerfData] = []
    servers: list[ServerSpec] = []

    for i in range(n_models):
        model_name = f"model-{i:05d}"
        ci = i % 2
        if ci == 0:  # Premium: tight SLOs (TPOT 24ms / TTFT 500ms class)
            itl = float(rng.uniform(20, 60))
            ttft = float(rng.uniform(400, 1200))
        else:  # Freemium
            itl = float(rng.uniform(100, 250))
            ttft = float(rng.uniform(1200, 3000))
        classes[ci].modelTargets.append(
            ModelTarget(model=model_name, slo_itl=itl, slo_ttft=ttft, slo_tps=0.0)
        )
        if perf_profiles is not None and model_name in perf_profiles:
            models.extend(perf_profiles[model_name])
        else:
            # random-init perf in the shape of fitted vLLM profiles; newer
            # accelerators are faster (lower alpha/beta)
            base_alpha = float(rng.uniform(6, 30))
            base_beta = float(rng.uniform(0.05, 0.6))
            base_gamma = float(rng.uniform(1, 12))
            base_delta = float(rng.uniform(2e-4, 6e-3))
            for j, a in enumerate(accs):
                f = 1.0 - 0.18 * j  # MI300X -> MI355X speedup ladder
                models.append(
                    ModelAcceleratorPerfData(
                        name=model_name,
                        acc=a.name,
                        accCount=int(rng.choice([1, 1, 1, 2, 4, 8])),
                        maxBatchSize=int(rng.choice([64, 128, 256])),
                        atTokens=int(rng.choice([512, 1024, 2048])),
                        decodeParms=DecodeParms(alpha=base_alpha * f, beta=base_beta * f),
                        prefillParms=PrefillParms(gamma=base_gamma * f, delta=base_delta * f),
                    )
                )
        cur_acc = accs[i % len(accs)].name
        servers.append(
            ServerSpec(
                name=f"srv-{i:05d}:bench",
                klass=classes[ci].name,
                model=model_name,
                keepAccelerator=False,
                minNumReplicas=min_num_replicas,
                maxBatchSize=0,
                currentAlloc=AllocationData(
                    accelerator=cur_acc,
                    numReplicas=1,
                    maxBatch=256,
                    cost=float(accs[i % len(accs)].cost),
                    load=ServerLoadSpec(
                        arrivalRate=float(rng.uniform(30, 600)),  # req/min
                        avgInTokens=int(rng.integers(64, 2048)),
                        avgOutTokens=int(rng.integers(32, 512)),
                    ),
                ),
            )
        )
    cap = capacity or {}
    return SystemSpec(
        accelerators=list(accs),
        models=models,
        serviceClasses=classes,
        servers=servers,
        optimizer=OptimizerSpec(unlimited=unlimited),
        capacity=[AcceleratorCount(type=t, count=c) for t, c in cap.items()],
    )


class PoissonTrace:
    """Bursty piecewise-rate Poisson arrival trace per server.

    Each server gets a base rate plus burst episodes; ``rates_at(step)``
    returns per-server arrival rates (req/min) — the Poisson-sampled count
    of arrivals over the measurement minute, like the collector's
    rate(vllm:request_success_total[1m])*60 estimate.
    """

    def __init__(self, n_servers: int, seed: int = 0, base_range=(30.0, 600.0)):
        self.rng = np.random.default_rng(seed)
        self.base = self.rng.uniform(base_range[0], base_range[1], size=n_servers)
        self.burst_phase = self.rng.uniform(0, 2 * np.pi, size=n_servers)
        self.burst_period = self.rng.integers(8, 40, size=n_servers)
        self.burst_gain = self.rng.uniform(1.5, 4.0, size=n_servers)

    def rates_at(self, step: int) -> np.ndarray:
        phase = 2 * np.pi * step / self.burst_period + self.burst_phase
        mult = 1.0 + (self.burst_gain - 1.0) * (np.sin(phase) > 0.6)
        lam = self.base * mult
        return self.rng.poisson(lam).astype(np.float64)
