"""VariantAutoscaling reconciler — the per-tick global optimization loop.

Mirrors internal/controller/variantautoscaling_controller.go:86-407:
read the three ConfigMaps, list active VAs, build SystemData via the
adapters, validate + collect Prometheus metrics per VA (continue-on-error so
one bad variant never blocks the fleet), run the batched analyzer+solver
(HIP sweep on MI355X; CPU golden fallback raises the ``SolverDegraded``
reason), then write DesiredOptimizedAlloc + conditions into each VA status
and emit the inferno_* gauges that HPA/KEDA consume. Actuation is
signal-only: the controller never patches Deployment replicas
(actuator.go:50-84).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Optional

from ..api import v1alpha1 as api
from ..core.system import System
from ..engine import SweepEngine
from ..parallel import ShardedSolver
from . import adapters, collector
from .k8s import ConflictError, Deployment, KubeClient
from .metrics import MetricsEmitter

CONFIGMAP_NAMESPACE = "workload-variant-autoscaler-system"
ACCELERATOR_CM = "accelerator-unit-costs"
SERVICE_CLASS_CM = "service-classes-config"
WVA_CONFIG_CM = "workload-variant-autoscaler-variantautoscaling-config"
DEFAULT_INTERVAL_SECONDS = 60.0


def _now_iso() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def _gpu_ready() -> bool:
    """A GPU re-probe may only swap the engine back when a device is visible
    AND the native library loads — otherwise the swapped-in engine fails on
    the next tick and churns through the fallback path repeatedly."""
    try:
        from ..engine.engine import _gpu_available

        if not _gpu_available():
            return False
        from ..ops.sweep import load_library

        load_library(allow_build=False)
        return True
    except Exception:
        return False


def parse_go_duration(s: str) -> float:
    """Parse a Go-style duration string ("60s", "1m30s", "500ms") to seconds."""
    import re

    if not s:
        raise ValueError("empty duration")
    units = {"h": 3600.0, "m": 60.0, "s": 1.0, "ms": 1e-3, "us": 1e-6, "ns": 1e-9}
    total = 0.0
    matched = False
    for num, unit in re.findall(r"([0-9]*\.?[0-9]+)(h|ms|us|ns|m|s)", s):
        total += float(num) * units[unit]
        matched = True
    if not matched:
        raise ValueError(f"invalid duration {s!r}")
    return total


@dataclass
class ReconcileResult:
    requeue_after: float = DEFAULT_INTERVAL_SECONDS
    processed: int = 0
    solver_backend: str = ""
    degraded: bool = False
    errors: list[str] = field(default_factory=list)
    duration_seconds: float = 0.0


class Actuator:
    """Signal-only actuation: gauges from real Deployment replica counts.

    Ref internal/actuator/actuator.go:28-84.
    """

    def __init__(self, kube: KubeClient, emitter: MetricsEmitter):
        self.kube = kube
        self.emitter = emitter

    def current_deployment_replicas(self, va: api.VariantAutoscaling) -> int:
        deploy = self.kube.get_deployment(va.namespace, va.name)
        if deploy is None:
            # fallback to the (possibly stale) VA status
            return va.status.currentAlloc.numReplicas
        if deploy.status_replicas >= 0:
            return deploy.status_replicas
        if deploy.replicas is not None:
            return deploy.replicas
        return 1

    def emit_metrics(self, va: api.VariantAutoscaling) -> None:
        if va.status.desiredOptimizedAlloc.numReplicas < 0:
            return
        current = self.current_deployment_replicas(va)
        desired = va.status.desiredOptimizedAlloc.numReplicas
        self.emitter.emit_replica_metrics(
            va.name,
            va.namespace,
            current,
            desired,
            va.status.desiredOptimizedAlloc.accelerator,
        )
        # scaling-direction counter (the reference registers this series but
        # never increments it — internal/metrics/metrics.go:84-101; here a
        # recommendation delta counts as one scaling operation)
        if desired > current:
            self.emitter.emit_replica_scaling(va.name, va.namespace, "up",
                                              "slo_optimization")
        elif desired < current:
            self.emitter.emit_replica_scaling(va.name, va.namespace, "down",
                                              "cost_optimization")


class Reconciler:
    def __init__(
        self,
        kube: KubeClient,
        prom: collector.PromAPI,
        emitter: MetricsEmitter,
        backend: str = "auto",
        configmap_namespace: str = CONFIGMAP_NAMESPACE,
        scale_to_zero: Optional[bool] = None,
    ):
        self.kube = kube
        self.prom = prom
        self.emitter = emitter
        self.actuator = Actuator(kube, emitter)
        self.engine = SweepEngine(backend=backend)
        self.solver = ShardedSolver(self.engine)
        self.configmap_namespace = configmap_namespace
        self.scale_to_zero = scale_to_zero
        self.last_result: Optional[ReconcileResult] = None
        # GPU->CPU degradation bookkeeping: after a GPU solver failure we run
        # on the CPU golden, but re-probe the GPU backend every
        # ``gpu_reprobe_interval`` reconciles so a transient HIP error does
        # not leave the controller degraded until restart.
        self._requested_backend = backend
        self._degraded_ticks = 0
        self.gpu_reprobe_interval = 10
        # thread-pool width for the I/O-bound per-VA collection phase; the
        # pool is created lazily and reused across reconciles (thread startup
        # per tick measured at >100ms)
        import os as _os

        self.collect_workers = int(_os.environ.get("WVA_COLLECT_WORKERS", "8"))
        self._collect_pool = None

    def _pool(self):
        if self._collect_pool is None:
            from concurrent.futures import ThreadPoolExecutor

            self._collect_pool = ThreadPoolExecutor(
                max_workers=self.collect_workers,
                thread_name_prefix="wva-collect",
            )
        return self._collect_pool

    # ------------------------------------------------------------------
    def read_interval(self) -> float:
        try:
            cm = self.kube.get_configmap(self.configmap_namespace, WVA_CONFIG_CM) or {}
        except Exception:
            # API unreachable: keep the default requeue cadence (the tick
            # itself will record the error; ref requeue-on-error semantics)
            return DEFAULT_INTERVAL_SECONDS
        interval = cm.get("GLOBAL_OPT_INTERVAL", "")
        if interval:
            try:
                return parse_go_duration(interval)
            except ValueError:
                pass
        return DEFAULT_INTERVAL_SECONDS

    def _read_accelerator_cm(self) -> Optional[dict[str, str]]:
        return self.kube.get_configmap(self.configmap_namespace, ACCELERATOR_CM)

    def _read_service_class_cm(self) -> Optional[dict[str, str]]:
        return self.kube.get_configmap(self.configmap_namespace, SERVICE_CLASS_CM)

    # ------------------------------------------------------------------
    def _update_status_with_retry(self, va: api.VariantAutoscaling,
                                  attempts: int = 3) -> None:
        """Status write with conflict-refetch-retry, the analogue of the
        reference's UpdateStatusWithBackoff (internal/utils/utils.go:91-104):
        on a 409, refetch the live object, graft our computed status onto the
        fresh resourceVersion and retry."""
        for i in range(attempts):
            try:
                self.kube.update_va_status(va)
                return
            except ConflictError:
                if i == attempts - 1:
                    raise
                getter = getattr(self.kube, "get_variantautoscaling", None)
                fresh = getter(va.namespace, va.name) if getter else None
                if fresh is None:
                    raise
                va.resourceVersion = fresh.resourceVersion

    def _maybe_reprobe_gpu(self) -> None:
        if not self.engine.degraded_from_gpu:
            return
        self._degraded_ticks += 1
        if self._degraded_ticks < self.gpu_reprobe_interval:
            return
        self._degraded_ticks = 0
        try:
            if not _gpu_ready():
                return
            probe = SweepEngine(backend="gpu")
            if probe.backend == "gpu":
                self.engine = probe
                self.solver = ShardedSolver(self.engine)
        except Exception:
            pass

    # ------------------------------------------------------------------
    def reconcile(self) -> ReconcileResult:
        t_start = time.perf_counter()
        self._maybe_reprobe_gpu()
        result = ReconcileResult(requeue_after=self.read_interval())

        # API-server errors on the tick's list/ConfigMap reads must degrade
        # to a failed tick, never crash the process (ref: controller-runtime
        # logs and requeues on client errors)
        try:
            accelerator_cm = self._read_accelerator_cm()
            if accelerator_cm is None:
                result.errors.append("unable to read accelerator configMap")
                return self._finish(result, t_start)
            service_class_cm = self._read_service_class_cm()
            if service_class_cm is None:
                result.errors.append("unable to read serviceclass configMap")
                return self._finish(result, t_start)

            vas = [va for va in self.kube.list_variantautoscalings()
                   if not va.deletionTimestamp]
        except Exception as e:
            result.errors.append(f"API server unreachable: {e}")
            return self._finish(result, t_start)
        if not vas:
            return self._finish(result, t_start)

        spec = adapters.create_system_data(accelerator_cm, service_class_cm)
        # Opt-in limited mode (beyond the reference, which forces Unlimited
        # and stubs inventory collection — collector.go:37-42): capacity from
        # the cluster's GPU node labels constrains the greedy solver.
        import os as _os

        if _os.environ.get("WVA_LIMITED_MODE") == "true":
            inventory = collector.collect_inventory_k8s(self.kube)
            capacity = adapters.capacity_from_inventory(spec, inventory)
            if capacity:
                spec.optimizer.unlimited = False
                spec.optimizer.saturationPolicy = _os.environ.get(
                    "WVA_SATURATION_POLICY", "None"
                )
                spec.capacity = capacity
        import json

        acc_costs: dict[str, float] = {}
        for key, val in accelerator_cm.items():
            try:
                acc_costs[key] = float(json.loads(val)["cost"])
            except (json.JSONDecodeError, KeyError, TypeError, ValueError):
                continue

        # ---- prepare phase (per-VA, continue on error) -------------------
        # Collection is I/O-bound (5 PromQL + k8s gets per VA, ref
        # controller.go:218-335 does them serially); here the per-VA network
        # phase fans out over a thread pool and only the shared-spec mutation
        # runs serially. Per-VA failures never block the fleet.
        def _collect(va: api.VariantAutoscaling):
            model_name = va.spec.modelID
            if not model_name:
                return None
            try:
                _, class_name = adapters.find_model_slo(service_class_cm, model_name)
            except adapters.AdapterError as e:
                return f"{va.name}: {e}"
            acc_name = va.labels.get(api.ACCELERATOR_LABEL, "")
            if acc_name not in acc_costs:
                return f"{va.name}: missing accelerator cost for {acc_name!r}"
            deploy = self.kube.get_deployment(va.namespace, va.name)
            if deploy is None:
                return f"{va.name}: deployment not found"
            if not any(r.get("uid") == deploy.uid for r in va.ownerReferences):
                self.kube.set_owner_reference(va, deploy)

            validation = collector.validate_metrics_availability(
                self.prom, model_name, deploy.namespace
            )
            if validation.available:
                api.set_condition(
                    va,
                    api.TYPE_METRICS_AVAILABLE,
                    "True",
                    validation.reason,
                    validation.message,
                )
            else:
                # metrics unavailable: log and skip (ref controller.go:305-316)
                return f"{va.name}: metrics unavailable ({validation.reason})"

            try:
                current_alloc = collector.add_metrics_to_opt_status(
                    va, deploy.namespace, deploy.replicas, acc_costs[acc_name], self.prom
                )
            except Exception as e:
                return f"{va.name}: metric collection failed: {e}"
            va.status.currentAlloc = current_alloc
            return (va, class_name)

        def _collect_safe(va: api.VariantAutoscaling):
            try:
                return _collect(va)
            except Exception as e:  # noqa: BLE001 - per-VA continue-on-error
                return f"{va.name}: prepare failed: {e}"

        if self.collect_workers > 1 and len(vas) > 1:
            collected = list(self._pool().map(_collect_safe, vas))
        else:
            collected = [_collect_safe(va) for va in vas]

        update_list: list[api.VariantAutoscaling] = []
        for va, item in zip(vas, collected):
            if item is None:
                continue
            if isinstance(item, str):
                result.errors.append(item)
                continue
            va, class_name = item
            # serial: these mutate the shared SystemSpec
            for profile in va.spec.modelProfile.accelerators:
                try:
                    adapters.add_model_accelerator_profile(spec, va.spec.modelID, profile)
                except adapters.AdapterError:
                    continue
            try:
                adapters.add_server_info(spec, va, class_name, self.scale_to_zero)
            except adapters.AdapterError as e:
                result.errors.append(f"{va.name}: bad server data: {e}")
                continue
            update_list.append(va)

        if not update_list:
            return self._finish(result, t_start)

        # ---- analyze + optimize (batched sweep) --------------------------
        system, opt_spec = System.from_spec(spec)
        for acc in system.accelerators.values():
            acc.calculate()
        try:
            shard_result = self.solver.solve(system, opt_spec)
            solution = shard_result.solution
            result.solver_backend = self.engine.backend
        except Exception as e:
            if self.engine.backend == "gpu":
                # GPU failure -> CPU reference fallback with SolverDegraded
                # (SURVEY.md section 5 failure-detection plan); periodically
                # re-probed by _maybe_reprobe_gpu
                self.engine = SweepEngine(backend="cpu")
                self.engine.degraded_from_gpu = True
                self.solver = ShardedSolver(self.engine)
                try:
                    shard_result = self.solver.solve(system, opt_spec)
                    solution = shard_result.solution
                    result.solver_backend = "cpu"
                    result.degraded = True
                except Exception as e2:
                    return self._fail_all(update_list, result, t_start, e2)
            else:
                return self._fail_all(update_list, result, t_start, e)

        if not solution:
            return self._fail_all(
                update_list, result, t_start,
                RuntimeError("no feasible allocations found for all variants"),
            )

        # ---- apply phase -------------------------------------------------
        now = _now_iso()
        for va in update_list:
            try:
                optimized = adapters.create_optimized_alloc(va.name, va.namespace, solution, now)
            except adapters.AdapterError:
                continue
            va.status.desiredOptimizedAlloc = optimized
            va.status.actuation.applied = False
            reason = api.REASON_OPTIMIZATION_SUCCEEDED
            message = (
                f"Optimization completed: {optimized.numReplicas} replicas on "
                f"{optimized.accelerator}"
            )
            if result.degraded:
                reason = api.REASON_SOLVER_DEGRADED
                message += " (GPU solver degraded to CPU reference)"
            api.set_condition(va, api.TYPE_OPTIMIZATION_READY, "True", reason, message)
            try:
                self.actuator.emit_metrics(va)
                va.status.actuation.applied = True
            except Exception as e:
                result.errors.append(f"{va.name}: metric emission failed: {e}")
            try:
                self._update_status_with_retry(va)
            except Exception as e:
                result.errors.append(f"{va.name}: status update failed: {e}")
                continue
            result.processed += 1

        return self._finish(result, t_start)

    # ------------------------------------------------------------------
    def _fail_all(self, update_list, result, t_start, err) -> ReconcileResult:
        for va in update_list:
            api.set_condition(
                va,
                api.TYPE_OPTIMIZATION_READY,
                "False",
                api.REASON_OPTIMIZATION_FAILED,
                f"Optimization failed: {err}",
            )
            try:
                self._update_status_with_retry(va)
            except Exception:
                pass
        result.errors.append(f"optimization failed: {err}")
        return self._finish(result, t_start)

    def _finish(self, result: ReconcileResult, t_start: float) -> ReconcileResult:
        result.duration_seconds = time.perf_counter() - t_start
        try:
            self.emitter.reconcile_latency.observe(result.duration_seconds)
        except Exception:
            pass
        self.last_result = result
        return result
