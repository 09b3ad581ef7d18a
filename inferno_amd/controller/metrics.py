"""Custom Prometheus metrics — schema-identical to the reference
(internal/metrics/metrics.go:20-126): inferno_replica_scaling_total counter
and inferno_{desired,current}_replicas / inferno_desired_ratio gauges with
labels {variant_name, namespace, accelerator_type} and the 0->N ratio
special case."""
from __future__ import annotations

from typing import Optional

from prometheus_client import Counter, Gauge, REGISTRY

from . import constants as c

_metrics: Optional["MetricsEmitter"] = None


class MetricsEmitter:
    def __init__(self, registry=REGISTRY):
        self.replica_scaling_total = Counter(
            c.INFERNO_REPLICA_SCALING_TOTAL.removesuffix("_total"),
            "Total number of replica scaling operations",
            [c.LABEL_VARIANT_NAME, c.LABEL_NAMESPACE, c.LABEL_DIRECTION, c.LABEL_REASON],
            registry=registry,
        )
        self.desired_replicas = Gauge(
            c.INFERNO_DESIRED_REPLICAS,
            "Desired number of replicas for each variant",
            [c.LABEL_VARIANT_NAME, c.LABEL_NAMESPACE, c.LABEL_ACCELERATOR_TYPE],
            registry=registry,
        )
        self.current_replicas = Gauge(
            c.INFERNO_CURRENT_REPLICAS,
            "Current number of replicas for each variant",
            [c.LABEL_VARIANT_NAME, c.LABEL_NAMESPACE, c.LABEL_ACCELERATOR_TYPE],
            registry=registry,
        )
        self.desired_ratio = Gauge(
            c.INFERNO_DESIRED_RATIO,
            "Ratio of the desired number of replicas and the current number of "
            "replicas for each variant",
            [c.LABEL_VARIANT_NAME, c.LABEL_NAMESPACE, c.LABEL_ACCELERATOR_TYPE],
            registry=registry,
        )
        # this build only: reconcile latency histogram (SURVEY.md section 5
        # "new build: keep a reconcile-latency histogram metric")
        from prometheus_client import Histogram

        self.reconcile_latency = Histogram(
            "inferno_reconcile_duration_seconds",
            "End-to-end reconcile latency",
            buckets=(0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1, 2.5,
                     5, 10, 30, 60),
            registry=registry,
        )

    def emit_replica_scaling(self, va_name: str, namespace: str, direction: str,
                             reason: str) -> None:
        self.replica_scaling_total.labels(va_name, namespace, direction, reason).inc()

    def emit_replica_metrics(self, va_name: str, namespace: str, current: int,
                             desired: int, accelerator_type: str) -> None:
        """Ref metrics.go:103-126 including the 0 -> N ratio convention."""
        labels = (va_name, namespace, accelerator_type)
        self.current_replicas.labels(*labels).set(float(current))
        self.desired_replicas.labels(*labels).set(float(desired))
        if current == 0:
            self.desired_ratio.labels(*labels).set(float(desired))
        else:
            self.desired_ratio.labels(*labels).set(float(desired) / float(current))


def init_metrics(registry=None) -> MetricsEmitter:
    global _metrics
    if _metrics is None:
        if registry is not None:
            _metrics = MetricsEmitter(registry)
        else:
            _metrics = MetricsEmitter()
    return _metrics
