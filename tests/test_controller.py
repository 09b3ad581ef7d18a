"""Controller-stack tests: collector query wire-format, availability/staleness,
adapters, conditions, metrics emission, and the full reconcile loop against
the in-memory kube + mock Prometheus (the reference's envtest/mock tier)."""
import json
import time

import pytest
from prometheus_client import CollectorRegistry

from inferno_amd.api import v1alpha1 as api
from inferno_amd.controller import adapters, collector
from inferno_amd.controller.collector import MockPromAPI, Sample
from inferno_amd.controller.k8s import Deployment, InMemoryKube
from inferno_amd.controller.metrics import MetricsEmitter
from inferno_amd.controller.reconciler import Reconciler, parse_go_duration

NS = "workload-variant-autoscaler-system"

ACCELERATOR_CM = {
    "A100": json.dumps({"device": "NVIDIA-A100-PCIE-80GB", "cost": "40.00"}),
    "MI300X": json.dumps({"device": "AMD-MI300X-192GB", "cost": "65.00"}),
    "MI355X": json.dumps({"device": "AMD-MI355X-288GB", "cost": "95.00"}),
}

SERVICE_CLASS_CM = {
    "premium.yaml": (
        "name: Premium\npriority: 1\ndata:\n"
        "  - model: default/default\n    slo-tpot: 24\n    slo-ttft: 500\n"
        "  - model: meta/llama0-70b\n    slo-tpot: 80\n    slo-ttft: 500\n"
    ),
    "freemium.yaml": (
        "name: Freemium\npriority: 10\ndata:\n"
        "  - model: ibm/granite-13b\n    slo-tpot: 200\n    slo-ttft: 2000\n"
    ),
}


def make_va(name="vllme-deploy", ns="default", model="default/default", acc="MI355X"):
    return api.VariantAutoscaling(
        name=name,
        namespace=ns,
        labels={api.ACCELERATOR_LABEL: acc},
        spec=api.VariantAutoscalingSpec(
            modelID=model,
            sloClassRef=api.ConfigMapKeyRef(name="service-classes-config", key="premium.yaml"),
            modelProfile=api.ModelProfile(
                accelerators=[
                    api.AcceleratorProfile(
                        acc=acc,
                        accCount=1,
                        perfParms=api.PerfParms(
                            decodeParms={"alpha": "20.58", "beta": "0.41"},
                            prefillParms={"gamma": "5.2", "delta": "0.1"},
                        ),
                        maxBatchSize=4,
                    )
                ]
            ),
        ),
    )


class TestQueryWireFormat:
    def test_arrival_query(self):
        q = collector.arrival_query("m/x", "ns1")
        assert q == 'sum(rate(vllm:request_success_total{model_name="m/x",namespace="ns1"}[1m]))'

    def test_ratio_queries(self):
        q = collector.ttft_query("m", "n")
        assert q == (
            'sum(rate(vllm:time_to_first_token_seconds_sum{model_name="m",namespace="n"}[1m]))'
            '/sum(rate(vllm:time_to_first_token_seconds_count{model_name="m",namespace="n"}[1m]))'
        )
        assert "vllm:time_per_output_token_seconds_sum" in collector.itl_query("m", "n")
        assert "vllm:request_prompt_tokens_sum" in collector.avg_prompt_tokens_query("m", "n")
        assert "vllm:request_generation_tokens_sum" in collector.avg_decode_tokens_query("m", "n")

    def test_fix_value(self):
        assert collector.fix_value(float("nan")) == 0.0
        assert collector.fix_value(float("inf")) == 0.0
        assert collector.fix_value(1.25) == 1.25


class TestAvailability:
    def test_available(self):
        prom = MockPromAPI()
        res = collector.validate_metrics_availability(prom, "m", "ns")
        assert res.available and res.reason == api.REASON_METRICS_FOUND

    def test_missing(self):
        prom = MockPromAPI(default_value=1.0)
        q1 = 'vllm:num_requests_running{model_name="m",namespace="ns"}'
        q2 = 'vllm:num_requests_running{model_name="m"}'
        prom.results = {q1: [], q2: []}
        res = collector.validate_metrics_availability(prom, "m", "ns")
        assert not res.available and res.reason == api.REASON_METRICS_MISSING

    def test_emulator_fallback_without_namespace(self):
        q1 = 'vllm:num_requests_running{model_name="m",namespace="ns"}'
        prom = MockPromAPI(results={q1: []})
        res = collector.validate_metrics_availability(prom, "m", "ns")
        assert res.available

    def test_stale(self):
        q1 = 'vllm:num_requests_running{model_name="m",namespace="ns"}'
        prom = MockPromAPI(results={q1: [Sample(1.0, time.time() - 400)]})
        res = collector.validate_metrics_availability(prom, "m", "ns")
        assert not res.available and res.reason == api.REASON_METRICS_STALE

    def test_prometheus_error(self):
        q1 = 'vllm:num_requests_running{model_name="m",namespace="ns"}'
        prom = MockPromAPI(errors={q1: RuntimeError("boom")})
        res = collector.validate_metrics_availability(prom, "m", "ns")
        assert not res.available and res.reason == api.REASON_PROMETHEUS_ERROR


class TestCollectStatus:
    def test_status_strings_and_units(self):
        va = make_va()
        prom = MockPromAPI(
            results={
                collector.arrival_query("default/default", "default"): [Sample(0.5, time.time())],
                collector.ttft_query("default/default", "default"): [Sample(0.015, time.time())],
                collector.itl_query("default/default", "default"): [Sample(0.007, time.time())],
                collector.avg_prompt_tokens_query("default/default", "default"): [
                    Sample(128.4, time.time())
                ],
                collector.avg_decode_tokens_query("default/default", "default"): [
                    Sample(64.6, time.time())
                ],
            }
        )
        alloc = collector.add_metrics_to_opt_status(va, "default", 3, 95.0, prom)
        assert alloc.accelerator == "MI355X"
        assert alloc.numReplicas == 3
        assert alloc.maxBatch == 256
        assert alloc.variantCost == "285.00"
        assert alloc.ttftAverage == "15.00"  # s -> ms, 2 decimals
        assert alloc.itlAverage == "7.00"
        assert alloc.load.arrivalRate == "30.00"  # req/s -> req/min
        assert alloc.load.avgInputTokens == "128.40"
        assert alloc.load.avgOutputTokens == "64.60"


class TestAdapters:
    def test_create_system_data(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        assert {a.name for a in spec.accelerators} == {"A100", "MI300X", "MI355X"}
        a100 = next(a for a in spec.accelerators if a.name == "A100")
        assert a100.type == "NVIDIA-A100-PCIE-80GB" and a100.cost == 40.0
        assert spec.optimizer.unlimited is True  # forced (utils.go:170-173)
        prem = next(c for c in spec.serviceClasses if c.name == "Premium")
        assert prem.priority == 1
        t = next(mt for mt in prem.modelTargets if mt.model == "default/default")
        assert t.slo_itl == 24 and t.slo_ttft == 500

    def test_bad_accelerator_entry_skipped(self):
        cm = dict(ACCELERATOR_CM)
        cm["BAD"] = "not json"
        spec = adapters.create_system_data(cm, SERVICE_CLASS_CM)
        assert "BAD" not in {a.name for a in spec.accelerators}

    def test_find_model_slo(self):
        entry, cls = adapters.find_model_slo(SERVICE_CLASS_CM, "ibm/granite-13b")
        assert cls == "Freemium" and entry.slo_tpot == 200
        with pytest.raises(adapters.AdapterError):
            adapters.find_model_slo(SERVICE_CLASS_CM, "nope")

    def test_add_model_accelerator_profile(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        va = make_va()
        adapters.add_model_accelerator_profile(
            spec, va.spec.modelID, va.spec.modelProfile.accelerators[0]
        )
        m = spec.models[0]
        assert m.acc == "MI355X" and m.decodeParms.alpha == pytest.approx(20.58)
        assert m.prefillParms.delta == pytest.approx(0.1)

    def test_add_model_accelerator_profile_rejects_short_parms(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        bad = api.AcceleratorProfile(acc="X", perfParms=api.PerfParms({"alpha": "1"}, {}))
        with pytest.raises(adapters.AdapterError):
            adapters.add_model_accelerator_profile(spec, "m", bad)

    def test_add_server_info(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        va = make_va()
        va.status.currentAlloc = api.Allocation(
            accelerator="MI355X",
            numReplicas=2,
            maxBatch=256,
            variantCost="190.00",
            itlAverage="9.50",
            ttftAverage="80.00",
            load=api.LoadProfile("120.00", "128.00", "64.00"),
        )
        adapters.add_server_info(spec, va, "Premium", scale_to_zero=False)
        s = spec.servers[0]
        assert s.name == "vllme-deploy:default"  # FullName
        assert s.keepAccelerator is True
        assert s.minNumReplicas == 1
        assert s.maxBatchSize == 4  # from the labeled accelerator profile
        assert s.currentAlloc.load.arrivalRate == pytest.approx(120.0)
        assert s.currentAlloc.cost == pytest.approx(190.0)

    def test_add_server_info_nan_guard(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        va = make_va()
        va.status.currentAlloc.load = api.LoadProfile("NaN", "garbage", "")
        adapters.add_server_info(spec, va, "Premium", scale_to_zero=False)
        assert spec.servers[0].currentAlloc.load.arrivalRate == 0.0

    def test_parse_go_duration(self):
        assert parse_go_duration("60s") == 60.0
        assert parse_go_duration("1m30s") == 90.0
        assert parse_go_duration("500ms") == 0.5
        with pytest.raises(ValueError):
            parse_go_duration("nope")


class TestConditions:
    def test_set_and_transition(self):
        va = make_va()
        api.set_condition(va, api.TYPE_METRICS_AVAILABLE, "True", "MetricsFound", "ok")
        c = api.get_condition(va, api.TYPE_METRICS_AVAILABLE)
        assert c is not None and c.status == "True"
        t1 = c.lastTransitionTime
        api.set_condition(va, api.TYPE_METRICS_AVAILABLE, "True", "MetricsFound", "ok2")
        assert api.get_condition(va, api.TYPE_METRICS_AVAILABLE).lastTransitionTime == t1
        api.set_condition(va, api.TYPE_METRICS_AVAILABLE, "False", "MetricsStale", "old")
        assert api.is_condition_false(va, api.TYPE_METRICS_AVAILABLE)

    def test_va_json_roundtrip(self):
        va = make_va()
        api.set_condition(va, api.TYPE_OPTIMIZATION_READY, "True", "OptimizationSucceeded", "m")
        doc = api.va_to_json(va)
        back = api.va_from_json(doc)
        assert api.va_to_json(back) == doc


class TestMetricsEmitter:
    def test_gauges_and_ratio(self):
        reg = CollectorRegistry()
        em = MetricsEmitter(registry=reg)
        em.emit_replica_metrics("v", "ns", current=2, desired=6, accelerator_type="MI355X")
        labels = {"variant_name": "v", "namespace": "ns", "accelerator_type": "MI355X"}
        assert reg.get_sample_value("inferno_desired_replicas", labels) == 6.0
        assert reg.get_sample_value("inferno_current_replicas", labels) == 2.0
        assert reg.get_sample_value("inferno_desired_ratio", labels) == 3.0

    def test_zero_current_ratio_special_case(self):
        reg = CollectorRegistry()
        em = MetricsEmitter(registry=reg)
        em.emit_replica_metrics("v", "ns", current=0, desired=4, accelerator_type="MI355X")
        labels = {"variant_name": "v", "namespace": "ns", "accelerator_type": "MI355X"}
        assert reg.get_sample_value("inferno_desired_ratio", labels) == 4.0

    def test_scaling_counter_name(self):
        reg = CollectorRegistry()
        em = MetricsEmitter(registry=reg)
        em.emit_replica_scaling("v", "ns", "up", "load")
        assert (
            reg.get_sample_value(
                "inferno_replica_scaling_total",
                {"variant_name": "v", "namespace": "ns", "direction": "up", "reason": "load"},
            )
            == 1.0
        )


def build_world(arrival_per_sec=2.0, model="default/default", acc="MI355X"):
    kube = InMemoryKube()
    kube.add_configmap(NS, "accelerator-unit-costs", ACCELERATOR_CM)
    kube.add_configmap(NS, "service-classes-config", SERVICE_CLASS_CM)
    kube.add_configmap(NS, "workload-variant-autoscaler-variantautoscaling-config",
                       {"GLOBAL_OPT_INTERVAL": "60s"})
    va = make_va(model=model, acc=acc)
    kube.add_va(va)
    kube.add_deployment(
        Deployment(name=va.name, namespace=va.namespace, replicas=1, status_replicas=1,
                   uid="uid-1")
    )
    now = time.time()
    prom = MockPromAPI(
        results={
            collector.arrival_query(model, "default"): [Sample(arrival_per_sec, now)],
            collector.ttft_query(model, "default"): [Sample(0.05, now)],
            collector.itl_query(model, "default"): [Sample(0.01, now)],
            collector.avg_prompt_tokens_query(model, "default"): [Sample(128, now)],
            collector.avg_decode_tokens_query(model, "default"): [Sample(64, now)],
        }
    )
    reg = CollectorRegistry()
    em = MetricsEmitter(registry=reg)
    rec = Reconciler(kube, prom, em, backend="cpu", scale_to_zero=False)
    return kube, prom, em, reg, rec


class TestReconcile:
    def test_full_loop_scale_out(self):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=6.0)
        result = rec.reconcile()
        assert result.processed == 1
        assert result.requeue_after == 60.0
        va = kube.vas[("default", "vllme-deploy")]
        des = va.status.desiredOptimizedAlloc
        assert des.accelerator == "MI355X"  # keepAccelerator restricts candidates
        assert des.numReplicas >= 1
        assert va.status.actuation.applied is True
        # conditions set
        assert api.is_condition_true(va, api.TYPE_METRICS_AVAILABLE)
        assert api.is_condition_true(va, api.TYPE_OPTIMIZATION_READY)
        # gauges emitted with deployment-backed current count
        labels = {"variant_name": "vllme-deploy", "namespace": "default",
                  "accelerator_type": "MI355X"}
        assert reg.get_sample_value("inferno_desired_replicas", labels) == float(
            des.numReplicas
        )
        assert reg.get_sample_value("inferno_current_replicas", labels) == 1.0
        # owner reference set from the deployment
        assert va.ownerReferences and va.ownerReferences[0]["uid"] == "uid-1"

    def test_high_load_needs_more_replicas(self):
        _, _, _, _, rec_low = build_world(arrival_per_sec=0.2)
        low = rec_low.reconcile()
        kube_hi, _, _, _, rec_hi = build_world(arrival_per_sec=50.0)
        hi = rec_hi.reconcile()
        assert hi.processed == 1 and low.processed == 1
        va_hi = kube_hi.vas[("default", "vllme-deploy")]
        assert va_hi.status.desiredOptimizedAlloc.numReplicas >= 2

    def test_metrics_unavailable_skips(self):
        kube, prom, em, reg, rec = build_world()
        q1 = 'vllm:num_requests_running{model_name="default/default",namespace="default"}'
        q2 = 'vllm:num_requests_running{model_name="default/default"}'
        prom.results[q1] = []
        prom.results[q2] = []
        result = rec.reconcile()
        assert result.processed == 0
        va = kube.vas[("default", "vllme-deploy")]
        assert va.status.desiredOptimizedAlloc.numReplicas == 0

    def test_missing_accelerator_cost_skips(self):
        kube, prom, em, reg, rec = build_world(acc="H100")
        result = rec.reconcile()
        assert result.processed == 0
        assert any("missing accelerator cost" in e for e in result.errors)

    def test_unknown_model_slo_skips(self):
        kube, prom, em, reg, rec = build_world(model="unknown/model")
        result = rec.reconcile()
        assert result.processed == 0

    def test_missing_configmap_aborts(self):
        kube, prom, em, reg, rec = build_world()
        del kube.configmaps[(NS, "accelerator-unit-costs")]
        result = rec.reconcile()
        assert result.processed == 0
        assert any("accelerator configMap" in e for e in result.errors)

    def test_deleted_va_filtered(self):
        kube, prom, em, reg, rec = build_world()
        kube.vas[("default", "vllme-deploy")].deletionTimestamp = "2026-01-01T00:00:00Z"
        result = rec.reconcile()
        assert result.processed == 0

    def test_interval_from_configmap(self):
        kube, prom, em, reg, rec = build_world()
        kube.add_configmap(NS, "workload-variant-autoscaler-variantautoscaling-config",
                           {"GLOBAL_OPT_INTERVAL": "30s"})
        assert rec.read_interval() == 30.0

    def test_status_update_recorded(self):
        kube, prom, em, reg, rec = build_world()
        rec.reconcile()
        assert len(kube.status_updates) == 1
        status = kube.status_updates[0]
        assert status["currentAlloc"]["load"]["arrivalRate"] == "120.00"
        assert status["desiredOptimizedAlloc"]["accelerator"] == "MI355X"


class TestPartialInfeasible:
    def test_infeasible_variant_skipped_others_proceed(self):
        """One variant with an impossible SLO is skipped; the rest of the
        fleet still gets optimized (per-VA continue-on-error)."""
        kube, prom, em, reg, rec = build_world(arrival_per_sec=2.0)
        # add a second VA with an SLO below its alpha (never satisfiable)
        bad = make_va(name="bad-deploy", model="meta/llama0-70b", acc="MI355X")
        bad.spec.modelProfile.accelerators[0].perfParms.decodeParms = {
            "alpha": "500.0", "beta": "1.0"  # ITL floor 500ms >> slo-tpot 80
        }
        kube.add_va(bad)
        kube.add_deployment(
            Deployment(name="bad-deploy", namespace="default", replicas=1,
                       status_replicas=1, uid="uid-bad")
        )
        now = time.time()
        for q in (
            collector.arrival_query("meta/llama0-70b", "default"),
            collector.avg_prompt_tokens_query("meta/llama0-70b", "default"),
            collector.avg_decode_tokens_query("meta/llama0-70b", "default"),
            collector.ttft_query("meta/llama0-70b", "default"),
            collector.itl_query("meta/llama0-70b", "default"),
        ):
            prom.results[q] = [Sample(2.0, now)]
        result = rec.reconcile()
        # the good variant got a decision; the infeasible one got none
        good = kube.vas[("default", "vllme-deploy")]
        assert good.status.desiredOptimizedAlloc.numReplicas >= 1
        bad_stored = kube.vas[("default", "bad-deploy")]
        assert bad_stored.status.desiredOptimizedAlloc.accelerator == ""
        assert result.processed == 1


class TestScalingCounter:
    def test_direction_counter_emitted_on_scale_out(self):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=50.0)
        rec.reconcile()
        va = kube.vas[("default", "vllme-deploy")]
        desired = va.status.desiredOptimizedAlloc.numReplicas
        assert desired > 1  # scale-out scenario
        val = reg.get_sample_value(
            "inferno_replica_scaling_total",
            {"variant_name": "vllme-deploy", "namespace": "default",
             "direction": "up", "reason": "slo_optimization"},
        )
        assert val == 1.0


class TestProbeServer:
    def test_probe_endpoints(self):
        import urllib.request

        from inferno_amd.controller.main import gpu_health_probe
        from inferno_amd.controller.serving import ProbeServer

        state = {"ready": False}
        srv = ProbeServer(0, state, bind="127.0.0.1")
        base = f"http://127.0.0.1:{srv.port}"

        def get(path):
            try:
                with urllib.request.urlopen(base + path, timeout=5) as r:
                    return r.status, r.read()
            except urllib.error.HTTPError as e:
                return e.code, b""

        assert get("/healthz")[0] == 200
        assert get("/readyz")[0] == 503  # not ready yet
        state["ready"] = True
        assert get("/readyz")[0] == 200
        code, body = get("/metrics")
        assert code == 200
        assert get("/nope")[0] == 404

        probe = gpu_health_probe()
        assert "gpu" in probe  # False in this container, True on an MI355X


class TestVaValidationEdges:
    """Mirrors the reference controller specs (variantautoscaling_controller_
    test.go:444-534): empty modelID / empty accelerator list / empty
    sloClassRef are handled gracefully without breaking other VAs."""

    def test_empty_model_id_skipped(self):
        kube, prom, em, reg, rec = build_world()
        bad = make_va(name="empty-model", model="")
        bad.spec.modelID = ""
        kube.add_va(bad)
        kube.add_deployment(Deployment(name="empty-model", namespace="default",
                                       replicas=1, status_replicas=1, uid="uid-em"))
        result = rec.reconcile()
        # the good VA still processes; the empty-modelID one is skipped
        assert result.processed == 1
        good = kube.vas[("default", "vllme-deploy")]
        assert api.is_condition_true(good, api.TYPE_OPTIMIZATION_READY)

    def test_empty_accelerator_list_does_not_break_others(self):
        kube, prom, em, reg, rec = build_world()
        bad = make_va(name="no-accs", model="default/default")
        bad.spec.modelProfile = api.ModelProfile(accelerators=[])
        kube.add_va(bad)
        kube.add_deployment(Deployment(name="no-accs", namespace="default",
                                       replicas=1, status_replicas=1, uid="uid-na"))
        result = rec.reconcile()
        good = kube.vas[("default", "vllme-deploy")]
        assert api.is_condition_true(good, api.TYPE_OPTIMIZATION_READY)
        assert result.processed >= 1

    def test_empty_slo_class_ref_uses_model_lookup(self):
        """The reconciler resolves SLOs by scanning the service-class
        ConfigMap for the model (FindModelSLO semantics) — an empty
        sloClassRef name does not crash the loop."""
        kube, prom, em, reg, rec = build_world()
        va = kube.vas[("default", "vllme-deploy")]
        va.spec.sloClassRef = api.ConfigMapKeyRef(name="", key="")
        result = rec.reconcile()
        assert result.processed == 1


class TestPrometheusConfigFromEnv:
    """Mirrors controller specs :253-410 — config resolution precedence and
    TLS defaults."""

    def test_missing_base_url_empty(self, monkeypatch):
        for k in ("PROMETHEUS_BASE_URL", "PROMETHEUS_BEARER_TOKEN",
                  "PROMETHEUS_CA_CERT_PATH", "PROMETHEUS_TOKEN_PATH"):
            monkeypatch.delenv(k, raising=False)
        cfg = collector.prometheus_config_from_env({})
        assert cfg["base_url"] == ""
        assert cfg["token"] is None
        assert cfg["insecure_skip_verify"] is False  # TLS verification default on
        assert cfg["allow_http"] is False

    def test_env_overrides_configmap(self, monkeypatch):
        monkeypatch.setenv("PROMETHEUS_BASE_URL", "https://env:9090")
        cfg = collector.prometheus_config_from_env(
            {"PROMETHEUS_BASE_URL": "https://cm:9090"})
        assert cfg["base_url"] == "https://env:9090"

    def test_configmap_fallback(self, monkeypatch):
        monkeypatch.delenv("PROMETHEUS_BASE_URL", raising=False)
        cfg = collector.prometheus_config_from_env(
            {"PROMETHEUS_BASE_URL": "https://cm:9090",
             "PROMETHEUS_TLS_INSECURE_SKIP_VERIFY": "true"})
        assert cfg["base_url"] == "https://cm:9090"
        assert cfg["insecure_skip_verify"] is True

    def test_https_enforced_by_client(self):
        with pytest.raises(ValueError):
            collector.PrometheusClient(base_url="http://insecure:9090")

    def test_http_allowed_when_opted_in(self):
        c = collector.PrometheusClient(base_url="http://dev:9090", allow_http=True)
        assert c is not None  # constructed without raising


class TestConflictRetry:
    """Status writes retry through resourceVersion conflicts
    (UpdateStatusWithBackoff analogue, internal/utils/utils.go:91-104)."""

    def test_conflict_then_refetch_succeeds(self):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=6.0)
        real_update = kube.update_va_status
        calls = {"n": 0}

        def flaky_update(va):
            calls["n"] += 1
            if calls["n"] == 1:
                # another writer bumped the object under us
                from inferno_amd.controller.k8s import ConflictError

                raise ConflictError("stale rv")
            real_update(va)

        kube.update_va_status = flaky_update
        result = rec.reconcile()
        assert result.processed == 1
        assert calls["n"] == 2  # one conflict, one successful retry
        assert not any("status update failed" in e for e in result.errors)

    def test_persistent_conflict_surfaces_error(self):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=6.0)

        def always_conflict(va):
            from inferno_amd.controller.k8s import ConflictError

            raise ConflictError("stale rv")

        kube.update_va_status = always_conflict
        result = rec.reconcile()
        assert result.processed == 0
        assert any("status update failed" in e for e in result.errors)


class TestGpuReprobe:
    """After a GPU solver failure the CPU fallback is periodically re-probed
    instead of degrading permanently."""

    def test_reprobe_counter_and_attempt(self, monkeypatch):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=2.0)
        rec.engine.degraded_from_gpu = True
        rec.gpu_reprobe_interval = 3
        probes = {"n": 0}

        import inferno_amd.controller.reconciler as rmod

        class FakeEngine:
            backend = "cpu"  # probe "fails": stays cpu -> no swap
            degraded_from_gpu = False

            def __init__(self, backend="auto", device="cuda"):
                probes["n"] += 1

        monkeypatch.setattr(rmod, "_gpu_ready", lambda: True)
        monkeypatch.setattr(rmod, "SweepEngine", FakeEngine)
        for _ in range(7):
            rec.reconcile()
        # ticks 3 and 6 triggered a probe
        assert probes["n"] == 2
        assert rec.engine.degraded_from_gpu is True  # probe failed, still degraded

    def test_successful_reprobe_restores_gpu(self, monkeypatch):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=2.0)
        rec.engine.degraded_from_gpu = True
        rec.gpu_reprobe_interval = 1

        import inferno_amd.controller.reconciler as rmod

        monkeypatch.setattr(rmod, "_gpu_ready", lambda: True)
        real_engine = type(rec.engine)

        class GpuOkEngine:
            def __new__(cls, backend="auto", device="cuda"):
                # build a real CPU engine but label it as a healthy GPU probe
                eng = real_engine(backend="cpu")
                eng.backend_probe = True
                eng.backend = "gpu"
                return eng

        monkeypatch.setattr(rmod, "SweepEngine", GpuOkEngine)
        rec._maybe_reprobe_gpu()
        assert rec.engine.backend == "gpu"
        assert rec.engine.degraded_from_gpu is False


class TestInventoryCollection:
    """Real CollectInventoryK8S (the reference's declared-TODO stub,
    collector.go:37-42) + opt-in limited mode fed by cluster GPU inventory."""

    def test_inventory_from_node_labels(self):
        from inferno_amd.controller.k8s import InMemoryKube, Node

        kube = InMemoryKube()
        kube.add_node(Node("n0", {"amd.com/gpu.count": "4",
                                  "amd.com/gpu.product": "MI355X",
                                  "amd.com/gpu.memory": "288GB"}))
        kube.add_node(Node("n1", {"amd.com/gpu.count": "4",
                                  "amd.com/gpu.product": "MI355X",
                                  "amd.com/gpu.memory": "288GB"}))
        kube.add_node(Node("n2", {"nvidia.com/gpu.count": "2",
                                  "nvidia.com/gpu.product": "A100",
                                  "nvidia.com/gpu.memory": "80GB"}))
        kube.add_node(Node("n3", {"amd.com/gpu.count": "bogus",
                                  "amd.com/gpu.product": "MI300X"}))
        kube.add_node(Node("n4", {}))  # no GPUs
        inv = collector.collect_inventory_k8s(kube)
        assert inv["amd.com"]["MI355X"] == {"count": 8, "memory": "288GB"}
        assert inv["nvidia.com"]["A100"]["count"] == 2
        assert "MI300X" not in inv.get("amd.com", {})

    def test_capacity_mapping_by_name_and_type(self):
        spec = adapters.create_system_data(ACCELERATOR_CM, SERVICE_CLASS_CM)
        inv = {"amd.com": {"MI355X": {"count": 8, "memory": "288GB"},
                           "AMD-MI300X-192GB": {"count": 4, "memory": "192GB"},
                           "UnknownChip": {"count": 9, "memory": "1GB"}}}
        caps = adapters.capacity_from_inventory(spec, inv)
        by_type = {c.type: c.count for c in caps}
        assert by_type["AMD-MI355X-288GB"] == 8  # matched by NAME
        assert by_type["AMD-MI300X-192GB"] == 4  # matched by TYPE string
        assert "UnknownChip" not in by_type

    def test_limited_mode_constrains_replicas(self, monkeypatch):
        from inferno_amd.controller.k8s import Node

        kube, prom, em, reg, rec = build_world(arrival_per_sec=50.0)
        # unlimited baseline needs >2 replicas under this load
        rec.reconcile()
        va = kube.vas[("default", "vllme-deploy")]
        unconstrained = va.status.desiredOptimizedAlloc.numReplicas
        assert unconstrained >= 2

        kube2, prom2, em2, reg2, rec2 = build_world(arrival_per_sec=50.0)
        kube2.add_node(Node("gpu-node", {"amd.com/gpu.count": "1",
                                         "amd.com/gpu.product": "MI355X",
                                         "amd.com/gpu.memory": "288GB"}))
        monkeypatch.setenv("WVA_LIMITED_MODE", "true")
        monkeypatch.setenv("WVA_SATURATION_POLICY", "PriorityExhaustive")
        rec2.reconcile()
        va2 = kube2.vas[("default", "vllme-deploy")]
        constrained = va2.status.desiredOptimizedAlloc.numReplicas
        assert 1 <= constrained <= 1  # capacity: 1 unit of MI355X
        assert constrained < unconstrained

    def test_limited_mode_no_nodes_falls_back_unlimited(self, monkeypatch):
        kube, prom, em, reg, rec = build_world(arrival_per_sec=6.0)
        monkeypatch.setenv("WVA_LIMITED_MODE", "true")
        result = rec.reconcile()
        assert result.processed == 1  # no inventory -> unlimited as before

    def test_no_gpu_no_swap(self):
        """On a box with no visible device, the re-probe never swaps a
        broken GPU engine in (would churn the fallback path)."""
        kube, prom, em, reg, rec = build_world(arrival_per_sec=2.0)
        rec.engine.degraded_from_gpu = True
        rec.gpu_reprobe_interval = 1
        rec._maybe_reprobe_gpu()  # container has no GPU -> _gpu_ready False
        assert rec.engine.backend == "cpu"
        assert rec.engine.degraded_from_gpu is True
