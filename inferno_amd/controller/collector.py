"""Collector — pulls vLLM load metrics from Prometheus.

PromQL strings are byte-compatible with the reference's collector
(internal/collector/collector.go:158-275): 5 queries per variant (arrival
rate, avg prompt/decode tokens, TTFT, ITL), availability validation with the
emulator fallback (no namespace label) and the 5-minute staleness cutoff
(collector.go:139-149). Floats land in the CR status as 2-decimal strings.
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Optional, Protocol

from ..api import v1alpha1 as api
from . import constants as c

STALENESS_CUTOFF_SECONDS = 5 * 60
# max batch size placeholder until collected from the server
# (ref collector.go:259 "maxBatch := 256")
DEFAULT_MAX_BATCH = 256


@dataclass
class Sample:
    value: float
    timestamp: float  # unix seconds


class PromAPI(Protocol):
    def query(self, promql: str) -> list[Sample]:  # pragma: no cover - protocol
        ...


class PrometheusClient:
    """Thin HTTP client for the Prometheus v1 query API (httpx).

    TLS/bearer behavior mirrors internal/utils/tls.go + prometheus_transport.go:
    HTTPS enforced unless ``allow_http`` (tests/emulator), optional CA bundle,
    client cert pair, and bearer token.
    """

    def __init__(
        self,
        base_url: str,
        token: Optional[str] = None,
        ca_cert: Optional[str] = None,
        client_cert: Optional[tuple[str, str]] = None,
        insecure_skip_verify: bool = False,
        allow_http: bool = False,
        timeout: float = 15.0,
    ):
        if not base_url:
            raise ValueError("missing Prometheus base URL")
        if not allow_http and not base_url.startswith("https://"):
            raise ValueError(
                f"Prometheus URL must use https:// (got {base_url!r})"
            )  # ref tls.go:63-68
        import httpx

        verify = False if insecure_skip_verify else (ca_cert or True)
        headers = {}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._client = httpx.Client(
            base_url=base_url, verify=verify, cert=client_cert, headers=headers,
            timeout=timeout,
        )

    def query(self, promql: str) -> list[Sample]:
        r = self._client.get("/api/v1/query", params={"query": promql})
        r.raise_for_status()
        body = r.json()
        if body.get("status") != "success":
            raise RuntimeError(f"prometheus query failed: {body}")
        data = body.get("data", {})
        if data.get("resultType") != "vector":
            return []
        out = []
        for item in data.get("result", []):
            ts, val = item.get("value", [0, "nan"])
            try:
                out.append(Sample(value=float(val), timestamp=float(ts)))
            except (TypeError, ValueError):
                continue
        return out

    def close(self) -> None:
        self._client.close()


def prometheus_config_from_env(cm: Optional[dict] = None) -> dict:
    """Prometheus client kwargs from env vars (the reference's env-name set,
    tls.go:98-117 ParsePrometheusConfigFromEnv) with ConfigMap fallback for
    each key (controller.go:541-580): PROMETHEUS_BASE_URL, _TLS_INSECURE_
    SKIP_VERIFY, _CA_CERT_PATH, _CLIENT_CERT_PATH/_CLIENT_KEY_PATH,
    _BEARER_TOKEN, _TOKEN_PATH."""
    import os

    cm = cm or {}

    def get(key: str, default: str = "") -> str:
        return os.environ.get(key) or cm.get(key, default)

    token = get("PROMETHEUS_BEARER_TOKEN")
    if not token:
        token_path = get("PROMETHEUS_TOKEN_PATH")
        if token_path and os.path.exists(token_path):
            with open(token_path) as f:
                token = f.read().strip()
    cert = get("PROMETHEUS_CLIENT_CERT_PATH")
    key = get("PROMETHEUS_CLIENT_KEY_PATH")
    return {
        "base_url": get("PROMETHEUS_BASE_URL"),
        "token": token or None,
        "ca_cert": get("PROMETHEUS_CA_CERT_PATH") or None,
        "client_cert": (cert, key) if cert and key else None,
        "insecure_skip_verify": get("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY") == "true",
        "allow_http": get("PROMETHEUS_ALLOW_HTTP") == "true",
    }


def validate_prometheus_api(prom: PromAPI, backoff=None) -> None:
    """Startup connectivity check with an "up" query under exponential
    backoff (ref internal/utils/utils.go:390-410 ValidatePrometheusAPI)."""
    from ..utils.backoff import PROMETHEUS_BACKOFF, retry_with_backoff

    retry_with_backoff(lambda: prom.query("up"), backoff or PROMETHEUS_BACKOFF)


def fix_value(x: float) -> float:
    """NaN/Inf -> 0 (ref collector.go:281-285)."""
    if math.isnan(x) or math.isinf(x):
        return 0.0
    return x


def _fmt2(x: float) -> str:
    """strconv.FormatFloat(x, 'f', 2, 32) equivalent (ref collector.go:268)."""
    return f"{float(x):.2f}"


# ---------------------------------------------------------------------------
# query builders (exact strings of collector.go:170-209)
# ---------------------------------------------------------------------------

def arrival_query(model: str, ns: str) -> str:
    return (
        f'sum(rate({c.VLLM_REQUEST_SUCCESS_TOTAL}{{{c.LABEL_MODEL_NAME}="{model}",'
        f'{c.LABEL_NAMESPACE}="{ns}"}}[1m]))'
    )


def _ratio_query(sum_metric: str, count_metric: str, model: str, ns: str) -> str:
    sel = f'{{{c.LABEL_MODEL_NAME}="{model}",{c.LABEL_NAMESPACE}="{ns}"}}'
    return f"sum(rate({sum_metric}{sel}[1m]))/sum(rate({count_metric}{sel}[1m]))"


def avg_prompt_tokens_query(model: str, ns: str) -> str:
    return _ratio_query(
        c.VLLM_REQUEST_PROMPT_TOKENS_SUM, c.VLLM_REQUEST_PROMPT_TOKENS_COUNT, model, ns
    )


def avg_decode_tokens_query(model: str, ns: str) -> str:
    return _ratio_query(
        c.VLLM_REQUEST_GENERATION_TOKENS_SUM, c.VLLM_REQUEST_GENERATION_TOKENS_COUNT, model, ns
    )


def ttft_query(model: str, ns: str) -> str:
    return _ratio_query(
        c.VLLM_TIME_TO_FIRST_TOKEN_SECONDS_SUM, c.VLLM_TIME_TO_FIRST_TOKEN_SECONDS_COUNT,
        model, ns,
    )


def itl_query(model: str, ns: str) -> str:
    return _ratio_query(
        c.VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_SUM, c.VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_COUNT,
        model, ns,
    )


# ---------------------------------------------------------------------------
# availability + metric collection
# ---------------------------------------------------------------------------

@dataclass
class MetricsValidationResult:
    available: bool
    reason: str
    message: str = ""


def validate_metrics_availability(
    prom: PromAPI, model_name: str, namespace: str, now: Optional[float] = None
) -> MetricsValidationResult:
    """Ref: collector.go:87-156 (namespace query, emulator fallback without
    namespace, then staleness check)."""
    now = now if now is not None else time.time()
    test_query = (
        f'{c.VLLM_NUM_REQUEST_RUNNING}{{{c.LABEL_MODEL_NAME}="{model_name}",'
        f'{c.LABEL_NAMESPACE}="{namespace}"}}'
    )
    try:
        vec = prom.query(test_query)
    except Exception as e:
        return MetricsValidationResult(
            False, api.REASON_PROMETHEUS_ERROR, f"Failed to query Prometheus: {e}"
        )
    if not vec:
        fallback = f'{c.VLLM_NUM_REQUEST_RUNNING}{{{c.LABEL_MODEL_NAME}="{model_name}"}}'
        try:
            vec = prom.query(fallback)
        except Exception as e:
            return MetricsValidationResult(
                False, api.REASON_PROMETHEUS_ERROR, f"Failed to query Prometheus: {e}"
            )
        if not vec:
            return MetricsValidationResult(
                False,
                api.REASON_METRICS_MISSING,
                f"No vLLM metrics found for model '{model_name}' in namespace "
                f"'{namespace}'. Check ServiceMonitor configuration and ensure vLLM "
                "pods are exposing /metrics endpoint",
            )
    for sample in vec:
        age = now - sample.timestamp
        if age > STALENESS_CUTOFF_SECONDS:
            return MetricsValidationResult(
                False,
                api.REASON_METRICS_STALE,
                f"vLLM metrics for model '{model_name}' are stale "
                f"(last update: {age:.0f}s ago). ServiceMonitor may not be scraping "
                "correctly.",
            )
    return MetricsValidationResult(
        True, api.REASON_METRICS_FOUND, "vLLM metrics are available and up-to-date"
    )


def _first_value(prom: PromAPI, q: str) -> float:
    vec = prom.query(q)
    if not vec:
        return 0.0
    return fix_value(vec[0].value)


def add_metrics_to_opt_status(
    va: api.VariantAutoscaling,
    deploy_namespace: str,
    deploy_replicas: int,
    accelerator_cost: float,
    prom: PromAPI,
) -> api.Allocation:
    """Build the CurrentAlloc status block from Prometheus + k8s data.

    Ref: collector.go:158-278 (unit conversions: arrival req/s -> req/min,
    TTFT/ITL s -> ms; cost = replicas * unit cost; accelerator from label).
    """
    model = va.spec.modelID
    ns = deploy_namespace

    arrival = _first_value(prom, arrival_query(model, ns)) * 60.0  # req/sec -> req/min
    avg_in = _first_value(prom, avg_prompt_tokens_query(model, ns))
    avg_out = _first_value(prom, avg_decode_tokens_query(model, ns))
    ttft_ms = _first_value(prom, ttft_query(model, ns)) * 1000.0
    itl_ms = _first_value(prom, itl_query(model, ns)) * 1000.0

    acc = va.labels.get(api.ACCELERATOR_LABEL, "")
    cost = float(deploy_replicas) * accelerator_cost

    return api.Allocation(
        accelerator=acc,
        numReplicas=int(deploy_replicas),
        maxBatch=DEFAULT_MAX_BATCH,
        variantCost=_fmt2(cost),
        ttftAverage=_fmt2(ttft_ms),
        itlAverage=_fmt2(itl_ms),
        load=api.LoadProfile(
            arrivalRate=_fmt2(arrival),
            avgInputTokens=_fmt2(avg_in),
            avgOutputTokens=_fmt2(avg_out),
        ),
    )


# GPU vendor label prefixes (ref collector.go:31-35)
GPU_VENDORS = ("nvidia.com", "amd.com", "intel.com")


def collect_inventory_k8s(kube=None) -> dict[str, dict[str, dict]]:
    """Cluster GPU inventory from node labels — the real implementation of
    the reference's declared-TODO stub (collector.go:37-42 CollectInventoryK8S,
    "will be properly implemented for limited mode").

    Scans nodes for the reference's emulated-GPU label convention
    (deploy/kind-emulator/setup.sh:120-133):
        {vendor}.com/gpu.count   "4"
        {vendor}.com/gpu.product "MI355X"
        {vendor}.com/gpu.memory  "288GB"
    Returns {vendor: {product: {"count": int, "memory": str}}} — the shape of
    the reference's map[string]map[string]AcceleratorModelInfo. Capacity for
    the limited-mode solver derives from this via
    ``adapters.capacity_from_inventory``."""
    if kube is None or not hasattr(kube, "list_nodes"):
        return {}
    inventory: dict[str, dict[str, dict]] = {}
    try:
        nodes = kube.list_nodes()
    except Exception:
        return {}
    for node in nodes:
        labels = node.labels or {}
        for vendor in GPU_VENDORS:
            count_s = labels.get(f"{vendor}/gpu.count", "")
            if not count_s:
                continue
            try:
                count = int(count_s)
            except ValueError:
                continue
            product = labels.get(f"{vendor}/gpu.product", "")
            memory = labels.get(f"{vendor}/gpu.memory", "")
            if count <= 0 or not product:
                continue
            entry = inventory.setdefault(vendor, {}).setdefault(
                product, {"count": 0, "memory": memory}
            )
            entry["count"] += count
            if memory and not entry["memory"]:
                entry["memory"] = memory
    return inventory


class MockPromAPI:
    """Test double mirroring the reference's MockPromAPI
    (test/utils/unitutils.go:137-159): query->samples map with a default
    non-empty vector so availability checks pass."""

    def __init__(self, results: Optional[dict[str, list[Sample]]] = None,
                 errors: Optional[dict[str, Exception]] = None,
                 default_value: float = 1.0):
        self.results = results or {}
        self.errors = errors or {}
        self.default_value = default_value
        self.queries: list[str] = []

    def query(self, promql: str) -> list[Sample]:
        self.queries.append(promql)
        if promql in self.errors:
            raise self.errors[promql]
        if promql in self.results:
            return self.results[promql]
        return [Sample(value=self.default_value, timestamp=time.time())]
