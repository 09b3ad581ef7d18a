"""Per-(server, accelerator) allocation sizing — the hot cell of the sweep.

CPU golden reference for HIP kernel K1/K2 ("allocate-sweep"); mirrors the
semantics of the reference's ``CreateAllocation`` (pkg/core/allocation.go:27-163),
``zeroLoadAllocation`` (:259-288) and ``TransitionPenalty`` (:291-300), with
the system passed explicitly instead of via the ``TheSystem`` singleton.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

from ..analyzer import (
    AnalyzerError,
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from ..config import (
    ACCEL_PENALTY_FACTOR,
    MAX_QUEUE_TO_BATCH_RATIO,
    AllocationData,
)


@dataclass
class Allocation:
    """Allocation of an accelerator to a server. Ref: allocation.go:13-24."""

    accelerator: str = ""
    num_replicas: int = 0
    batch_size: int = 0
    cost: float = 0.0
    value: float = 0.0
    itl: float = 0.0  # expected avg token decode time (msec)
    ttft: float = 0.0  # expected avg queueing + prefill time (msec)
    rho: float = 0.0
    max_arrv_rate_per_replica: float = 0.0  # req/msec

    @property
    def max_rpm(self) -> float:
        """Max requests/min per all replicas... per replica (ref allocation.go:224-226)."""
        return self.max_arrv_rate_per_replica * 1000.0 * 60.0

    def is_saturated(self, total_rate_rpm: float) -> bool:
        """Ref: allocation.go:254-256 (total rate in req/min)."""
        return total_rate_rpm > float(self.num_replicas) * self.max_rpm

    def transition_penalty(self, b: "Allocation") -> float:
        """Penalty of moving from this allocation to b. Ref: allocation.go:291-300."""
        if self.accelerator == b.accelerator:
            if self.num_replicas == b.num_replicas:
                return 0.0
            return b.cost - self.cost
        return ACCEL_PENALTY_FACTOR * (self.cost + b.cost) + (b.cost - self.cost)

    def to_data(self) -> AllocationData:
        return AllocationData(
            accelerator=self.accelerator,
            numReplicas=self.num_replicas,
            maxBatch=self.batch_size,
            cost=self.cost,
            itlAverage=self.itl,
            ttftAverage=self.ttft,
        )

    def clone(self) -> "Allocation":
        return Allocation(
            accelerator=self.accelerator,
            num_replicas=self.num_replicas,
            batch_size=self.batch_size,
            cost=self.cost,
            value=self.value,
            itl=self.itl,
            ttft=self.ttft,
            rho=self.rho,
            max_arrv_rate_per_replica=self.max_arrv_rate_per_replica,
        )


def allocation_from_data(data: AllocationData) -> Allocation:
    """Ref: allocation.go:349-358 AllocationFromData."""
    return Allocation(
        accelerator=data.accelerator,
        num_replicas=data.numReplicas,
        batch_size=data.maxBatch,
        cost=data.cost,
        itl=data.itlAverage,
        ttft=data.ttftAverage,
    )


@dataclass
class AllocationDiff:
    """Orchestration delta between two allocations. Ref: allocation.go:366-409."""

    old_accelerator: str = "none"
    new_accelerator: str = "none"
    old_num_replicas: int = 0
    new_num_replicas: int = 0
    cost_diff: float = 0.0


def create_allocation_diff(a: Optional[Allocation], b: Optional[Allocation]) -> Optional[AllocationDiff]:
    if a is None and b is None:
        return None
    d = AllocationDiff()
    if a is not None:
        d.old_accelerator = a.accelerator
        d.old_num_replicas = a.num_replicas
    if b is not None:
        d.new_accelerator = b.accelerator
        d.new_num_replicas = b.num_replicas
    d.cost_diff = (b.cost if b else 0.0) - (a.cost if a else 0.0)
    return d


def _zero_load_allocation(server, model, acc, perf) -> Allocation:
    """Allocation when there is no traffic. Ref: allocation.go:259-288."""
    num_replicas = server.min_num_replicas
    if num_replicas == 0:
        return Allocation()  # empty allocation, value/cost 0

    max_batch = perf.maxBatchSize
    if server.max_batch_size > 0:
        max_batch = server.max_batch_size
    total_instances = model.get_num_instances(acc.name) * num_replicas
    cost = acc.cost * float(total_instances)

    decode_time = perf.decodeParms.alpha + perf.decodeParms.beta
    max_decode_time = perf.decodeParms.alpha + perf.decodeParms.beta * float(max_batch)
    prefill_time = perf.prefillParms.gamma + perf.prefillParms.delta
    max_serv_time = prefill_time + max_decode_time
    max_arrv = float(max_batch) / max_serv_time if max_serv_time > 0 else 0.0

    alloc = Allocation(
        accelerator=acc.name,
        num_replicas=num_replicas,
        batch_size=max_batch,
        cost=cost,
        itl=decode_time,
        ttft=prefill_time,
        rho=0.0,
        max_arrv_rate_per_replica=max_arrv,
    )
    alloc.value = alloc.cost
    return alloc


def scale_allocation(system, alloc: Allocation, server_name: str):
    """Re-size this allocation's accelerator for the server's current load;
    returns (new allocation, replica delta). Ref: allocation.go:166-190 Scale."""
    server = system.servers.get(server_name)
    if server is None or server.load is None:
        return None, 0
    if system.accelerators.get(alloc.accelerator) is None:
        return None, 0
    new = create_allocation(system, server_name, alloc.accelerator)
    if new is None:
        return None, 0
    return new, new.num_replicas - alloc.num_replicas


def reallocate(system, server_name: str):
    """Min-value allocation across all accelerators; returns (allocation,
    accelerator name) or (None, ""). Ref: allocation.go:192-207 ReAllocate."""
    min_alloc: Optional[Allocation] = None
    for g_name in sorted(system.accelerators):
        alloc = create_allocation(system, server_name, g_name)
        if alloc is not None and (min_alloc is None or alloc.value < min_alloc.value):
            min_alloc = alloc
    if min_alloc is None:
        return None, ""
    return min_alloc, min_alloc.accelerator


def create_allocation(system, server_name: str, acc_name: str) -> Optional[Allocation]:
    """Size an accelerator for a server; None when infeasible.

    Mirrors CreateAllocation (pkg/core/allocation.go:27-163):
      1. resolve server/accelerator/model-perf/SLO-target, bail on any miss;
      2. zero traffic -> zero-load allocation;
      3. N from override or ``perf.maxBatchSize*atTokens/K`` (K=avg out tokens);
      4. queue analyzer sized at max rate meeting SLO targets -> rate*;
      5. replicas = ceil(totalRate / rate*), >= minNumReplicas;
      6. cost = acc.cost * numInstances * replicas;
      7. re-analyze at per-replica rate for expected ITL/TTFT/rho.
    """
    acc = system.accelerators.get(acc_name)
    if acc is None:
        return None
    server = system.servers.get(server_name)
    if server is None:
        return None
    load = server.load
    if load is None or load.arrivalRate < 0 or load.avgInTokens < 0 or load.avgOutTokens < 0:
        return None
    model = system.models.get(server.model_name)
    if model is None:
        return None
    perf = model.get_perf_data(acc_name)
    if perf is None:
        return None
    svc = system.service_classes.get(server.service_class_name)
    if svc is None:
        return None
    target = svc.model_target(server.model_name)
    if target is None:
        return None

    # zero traffic
    if load.arrivalRate == 0 or load.avgOutTokens == 0:
        return _zero_load_allocation(server, model, acc, perf)

    K = load.avgOutTokens
    if server.max_batch_size > 0:
        N = server.max_batch_size
    else:
        N = max(perf.maxBatchSize * perf.atTokens // K, 1)
    # N is uncapped, matching the reference (allocation.go:80-86); the GPU
    # sweep spills chain geometry for N > 8192 to global memory
    max_queue = N * MAX_QUEUE_TO_BATCH_RATIO

    cfg = Configuration(
        max_batch_size=N,
        max_queue_size=max_queue,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=perf.prefillParms.gamma, delta=perf.prefillParms.delta),
            decode=DecodeParms(alpha=perf.decodeParms.alpha, beta=perf.decodeParms.beta),
        ),
    )
    req = RequestSize(avg_input_tokens=load.avgInTokens, avg_output_tokens=K)
    try:
        if getattr(system, "analyzer_mode", "mm1k") == "mg1":
            from ..analyzer.mg1 import MG1QueueEvaluator

            qa = MG1QueueEvaluator(cfg, req, cv2=getattr(system, "analyzer_cv2", 1.0))
        else:
            qa = QueueAnalyzer(cfg, req)
    except AnalyzerError:
        return None

    targets = TargetPerf(
        target_ttft=target.ttft, target_itl=target.itl, target_tps=target.tps
    )
    try:
        _, metrics, _ = qa.size(targets)
    except AnalyzerError:
        return None
    rate_star = metrics.throughput  # req/sec

    if target.tps == 0:
        total_rate = load.arrivalRate / 60.0  # req/min -> req/sec
    else:
        total_rate = target.tps / float(K)
    num_replicas = int(math.ceil(total_rate / rate_star))
    num_replicas = max(num_replicas, server.min_num_replicas)

    total_instances = model.get_num_instances(acc_name) * num_replicas
    cost = acc.cost * float(total_instances)

    rate = total_rate / float(num_replicas)
    try:
        metrics = qa.analyze(rate)
    except AnalyzerError:
        return None

    alloc = Allocation(
        accelerator=acc_name,
        num_replicas=num_replicas,
        batch_size=N,
        cost=cost,
        itl=metrics.avg_token_time,
        ttft=metrics.avg_wait_time + metrics.avg_prefill_time,
        rho=metrics.rho,
        max_arrv_rate_per_replica=rate_star / 1000.0,
    )
    alloc.value = alloc.cost
    return alloc
