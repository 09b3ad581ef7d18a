"""Prometheus exposition of the exact ``vllm:*`` series the autoscaler's
collector scrapes (ref tools/vllm-emulator/metrics.py:7-80 and
internal/constants/metrics.go:7-47)."""
from __future__ import annotations

from prometheus_client import CollectorRegistry
from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily

from .sim import VLLMSim


class VllmMetricsCollector:
    """Custom collector so sum/count pairs keep their exact wire names."""

    def __init__(self, sim: VLLMSim, model_name: str, namespace: str = ""):
        self.sim = sim
        self.model_name = model_name
        self.namespace = namespace

    def _labels(self):
        names = ["model_name"] + (["namespace"] if self.namespace else [])
        values = [self.model_name] + ([self.namespace] if self.namespace else [])
        return names, values

    def collect(self):
        names, values = self._labels()
        sim = self.sim

        g = GaugeMetricFamily("vllm:num_requests_running",
                              "Number of requests currently running", labels=names)
        g.add_metric(values, sim.num_requests_running)
        yield g
        w = GaugeMetricFamily("vllm:num_requests_waiting",
                              "Number of requests waiting", labels=names)
        w.add_metric(values, sim.num_requests_waiting)
        yield w

        c = CounterMetricFamily("vllm:request_success", "Successful requests", labels=names)
        c.add_metric(values, sim.success_total)
        yield c

        pairs = [
            ("vllm:request_prompt_tokens", sim.prompt_tokens_sum, sim.prompt_tokens_count),
            ("vllm:request_generation_tokens", sim.generation_tokens_sum,
             sim.generation_tokens_count),
            ("vllm:time_to_first_token_seconds", sim.ttft_sum_s, sim.ttft_count),
            ("vllm:time_per_output_token_seconds", sim.tpot_sum_s, sim.tpot_count),
        ]
        for base, s, cnt in pairs:
            gs = GaugeMetricFamily(f"{base}_sum", f"{base} sum", labels=names)
            gs.add_metric(values, s)
            yield gs
            gc = GaugeMetricFamily(f"{base}_count", f"{base} count", labels=names)
            gc.add_metric(values, cnt)
            yield gc

        u = GaugeMetricFamily("vllm:gpu_cache_usage_perc", "KV cache usage fraction",
                              labels=names)
        u.add_metric(values, sim.device.utilization)
        yield u


def make_registry(sim: VLLMSim, model_name: str, namespace: str = "") -> CollectorRegistry:
    registry = CollectorRegistry()
    registry.register(VllmMetricsCollector(sim, model_name, namespace))
    return registry
