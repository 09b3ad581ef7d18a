"""HttpKube / PrometheusClient / LeaderElector against httpx MockTransport —
validates the REST paths, merge-patch bodies, watch stream parsing, lease
acquire/renew/steal logic and Prometheus config plumbing without a cluster."""
import json

import httpx
import pytest

from inferno_amd.api import v1alpha1 as api
from inferno_amd.controller.collector import (
    PrometheusClient,
    prometheus_config_from_env,
    validate_prometheus_api,
)
from inferno_amd.controller.k8s import Deployment, HttpKube
from inferno_amd.controller.leader import LeaderElector
from inferno_amd.utils.backoff import Backoff


def make_kube(handler):
    kube = HttpKube.__new__(HttpKube)
    kube._client = httpx.Client(
        base_url="https://kube.test", transport=httpx.MockTransport(handler)
    )
    return kube


class TestHttpKube:
    def test_list_vas(self):
        def handler(req):
            assert req.url.path == "/apis/llmd.ai/v1alpha1/variantautoscalings"
            return httpx.Response(
                200,
                json={
                    "items": [
                        {
                            "metadata": {"name": "v1", "namespace": "ns"},
                            "spec": {"modelID": "m"},
                        }
                    ]
                },
            )

        vas = make_kube(handler).list_variantautoscalings()
        assert len(vas) == 1 and vas[0].name == "v1" and vas[0].spec.modelID == "m"

    def test_get_configmap_and_404(self):
        def handler(req):
            if req.url.path.endswith("/configmaps/found"):
                return httpx.Response(200, json={"data": {"k": "v"}})
            return httpx.Response(404)

        kube = make_kube(handler)
        assert kube.get_configmap("ns", "found") == {"k": "v"}
        assert kube.get_configmap("ns", "missing") is None

    def test_get_deployment(self):
        def handler(req):
            assert req.url.path == "/apis/apps/v1/namespaces/ns/deployments/d"
            return httpx.Response(
                200,
                json={
                    "metadata": {"name": "d", "namespace": "ns", "uid": "u1"},
                    "spec": {"replicas": 3},
                    "status": {"replicas": 2},
                },
            )

        d = make_kube(handler).get_deployment("ns", "d")
        assert d.replicas == 3 and d.status_replicas == 2 and d.uid == "u1"

    def test_status_update_merge_patch(self):
        seen = {}

        def handler(req):
            seen["path"] = req.url.path
            seen["content_type"] = req.headers["content-type"]
            seen["body"] = json.loads(req.content)
            return httpx.Response(200, json={})

        kube = make_kube(handler)
        va = api.VariantAutoscaling(name="v1", namespace="ns")
        va.status.desiredOptimizedAlloc = api.OptimizedAlloc(
            accelerator="MI355X", numReplicas=4
        )
        kube.update_va_status(va)
        assert seen["path"] == "/apis/llmd.ai/v1alpha1/namespaces/ns/variantautoscalings/v1/status"
        assert seen["content_type"] == "application/merge-patch+json"
        assert seen["body"]["status"]["desiredOptimizedAlloc"]["numReplicas"] == 4

    def test_owner_reference_patch(self):
        seen = {}

        def handler(req):
            seen["body"] = json.loads(req.content)
            return httpx.Response(200, json={})

        kube = make_kube(handler)
        va = api.VariantAutoscaling(name="v1", namespace="ns")
        kube.set_owner_reference(va, Deployment(name="d", namespace="ns", uid="u9"))
        refs = seen["body"]["metadata"]["ownerReferences"]
        assert refs[0]["kind"] == "Deployment" and refs[0]["uid"] == "u9"
        assert va.ownerReferences == refs

    def test_watch_stream_parsing(self):
        lines = [
            json.dumps({"type": "ADDED", "object": {
                "kind": "VariantAutoscaling",
                "metadata": {"name": "w1", "namespace": "ns"}, "spec": {"modelID": "m"}}}),
            json.dumps({"type": "MODIFIED", "object": {
                "kind": "VariantAutoscaling",
                "metadata": {"name": "w1", "namespace": "ns"}}}),
            "not json",
        ]

        def handler(req):
            assert req.url.params["watch"] == "1"
            return httpx.Response(200, text="\n".join(lines))

        events = list(make_kube(handler).watch_events(timeout_seconds=1))
        assert [e[0] for e in events] == ["ADDED", "MODIFIED"]
        assert events[0][1].name == "w1"


class TestLeaderElector:
    def _kube_with(self, handler):
        return make_kube(handler)

    def test_acquire_when_absent(self):
        state = {"created": None}

        def handler(req):
            if req.method == "GET":
                return httpx.Response(404)
            if req.method == "POST":
                state["created"] = json.loads(req.content)
                return httpx.Response(201, json=state["created"])
            return httpx.Response(500)

        el = LeaderElector(self._kube_with(handler), "lease", "ns", "me")
        assert el.try_acquire() is True
        assert state["created"]["spec"]["holderIdentity"] == "me"

    def test_blocked_by_fresh_holder(self):
        from datetime import datetime, timezone

        now = datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%f0Z")

        def handler(req):
            return httpx.Response(200, json={
                "spec": {"holderIdentity": "other", "renewTime": now,
                         "leaseDurationSeconds": 15}})

        el = LeaderElector(self._kube_with(handler), "lease", "ns", "me")
        assert el.try_acquire() is False

    def test_steals_expired_lease(self):
        def handler(req):
            if req.method == "GET":
                return httpx.Response(200, json={
                    "spec": {"holderIdentity": "other",
                             "renewTime": "2020-01-01T00:00:00.0Z",
                             "leaseDurationSeconds": 15}})
            if req.method == "PUT":
                return httpx.Response(200, json={})
            return httpx.Response(500)

        el = LeaderElector(self._kube_with(handler), "lease", "ns", "me")
        assert el.try_acquire() is True


class TestPrometheusClient:
    def _client(self, handler, **kw):
        c = PrometheusClient.__new__(PrometheusClient)
        c._client = httpx.Client(
            base_url="https://prom.test", transport=httpx.MockTransport(handler)
        )
        return c

    def test_query_vector(self):
        def handler(req):
            assert req.url.path == "/api/v1/query"
            return httpx.Response(200, json={
                "status": "success",
                "data": {"resultType": "vector",
                         "result": [{"value": [1700000000.5, "2.25"]}]}})

        samples = self._client(handler).query("up")
        assert samples[0].value == 2.25 and samples[0].timestamp == 1700000000.5

    def test_https_enforced(self):
        with pytest.raises(ValueError):
            PrometheusClient("http://insecure:9090")

    def test_validate_with_backoff_retries(self):
        calls = {"n": 0}

        def handler(req):
            calls["n"] += 1
            if calls["n"] < 3:
                return httpx.Response(500)
            return httpx.Response(200, json={
                "status": "success", "data": {"resultType": "vector", "result": []}})

        prom = self._client(handler)
        validate_prometheus_api(prom, Backoff(duration=0.01, factor=1.5, steps=5))
        assert calls["n"] == 3

    def test_config_from_env(self, monkeypatch):
        monkeypatch.setenv("PROMETHEUS_BASE_URL", "https://p:9090")
        monkeypatch.setenv("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY", "true")
        monkeypatch.setenv("PROMETHEUS_BEARER_TOKEN", "tok")
        cfg = prometheus_config_from_env()
        assert cfg["base_url"] == "https://p:9090"
        assert cfg["insecure_skip_verify"] is True
        assert cfg["token"] == "tok"

    def test_config_cm_fallback(self, monkeypatch):
        for k in ("PROMETHEUS_BASE_URL", "PROMETHEUS_BEARER_TOKEN",
                  "PROMETHEUS_TLS_INSECURE_SKIP_VERIFY"):
            monkeypatch.delenv(k, raising=False)
        cfg = prometheus_config_from_env({"PROMETHEUS_BASE_URL": "https://cm:9090"})
        assert cfg["base_url"] == "https://cm:9090"


class TestConfigMapWatch:
    def test_filters_to_controller_configmaps(self):
        lines = [
            json.dumps({"type": "MODIFIED", "object": {
                "metadata": {"name": "service-classes-config",
                             "resourceVersion": "5"}}}),
            json.dumps({"type": "MODIFIED", "object": {
                "metadata": {"name": "unrelated-cm", "resourceVersion": "6"}}}),
            json.dumps({"type": "ADDED", "object": {
                "metadata": {"name": "accelerator-unit-costs",
                             "resourceVersion": "7"}}}),
        ]

        def handler(req):
            assert req.url.path == "/api/v1/namespaces/ns/configmaps"
            return httpx.Response(200, text="\n".join(lines))

        events = list(make_kube(handler).watch_configmap_events(
            "ns", {"service-classes-config", "accelerator-unit-costs"},
            timeout_seconds=1))
        assert events == [("MODIFIED", "service-classes-config", "5"),
                          ("ADDED", "accelerator-unit-costs", "7")]

    def test_resource_version_param_passed(self):
        def handler(req):
            assert req.url.params.get("resourceVersion") == "41"
            return httpx.Response(200, text="")

        list(make_kube(handler).watch_configmap_events(
            "ns", {"x"}, timeout_seconds=1, resource_version="41"))


class TestLeaderFailClosed:
    """Leader election must fail CLOSED on API errors (ADVICE r1 medium;
    ref cmd/main.go:201-219 controller-runtime semantics)."""

    def test_api_error_drops_leadership(self):
        def handler(req):
            return httpx.Response(500, json={})

        el = LeaderElector(make_kube(handler), "lease", "ns", "me")
        assert el.try_acquire() is False

    def test_unreachable_api_drops_leadership(self):
        def handler(req):
            raise httpx.ConnectError("coordination API down")

        el = LeaderElector(make_kube(handler), "lease", "ns", "me")
        assert el.try_acquire() is False

    def test_renew_conflict_loses_leadership(self):
        """A 409 on the renew PUT (concurrent takeover with a newer
        resourceVersion) must drop leadership."""
        def handler(req):
            if req.method == "GET":
                return httpx.Response(200, json={
                    "metadata": {"resourceVersion": "5"},
                    "spec": {"holderIdentity": "me",
                             "renewTime": "2020-01-01T00:00:00.0Z",
                             "leaseDurationSeconds": 15}})
            if req.method == "PUT":
                return httpx.Response(409, json={})
            return httpx.Response(500)

        el = LeaderElector(make_kube(handler), "lease", "ns", "me")
        assert el.try_acquire() is False

    def test_error_then_recovery_reacquires(self):
        state = {"fail": True}

        def handler(req):
            if state["fail"]:
                return httpx.Response(503)
            if req.method == "GET":
                return httpx.Response(404)
            if req.method == "POST":
                return httpx.Response(201, json={})
            return httpx.Response(500)

        el = LeaderElector(make_kube(handler), "lease", "ns", "me")
        assert el.try_acquire() is False
        state["fail"] = False
        assert el.try_acquire() is True

    def test_create_conflict_not_leader(self):
        """POST 409 (another replica created the Lease first) => not leader."""
        def handler(req):
            if req.method == "GET":
                return httpx.Response(404)
            if req.method == "POST":
                return httpx.Response(409, json={})
            return httpx.Response(500)

        el = LeaderElector(make_kube(handler), "lease", "ns", "me")
        assert el.try_acquire() is False


class TestStatusOptimisticConcurrency:
    """resourceVersion carried in status patches; 409 surfaces as
    ConflictError (ref internal/utils/utils.go:91-104)."""

    def test_patch_carries_resource_version(self):
        seen = {}

        def handler(req):
            seen["body"] = json.loads(req.content)
            return httpx.Response(200, json={"metadata": {"resourceVersion": "43"}})

        kube = make_kube(handler)
        va = api.VariantAutoscaling(name="v1", namespace="ns", resourceVersion="42")
        kube.update_va_status(va)
        assert seen["body"]["metadata"]["resourceVersion"] == "42"
        # server-returned rv adopted for the next write
        assert va.resourceVersion == "43"

    def test_conflict_raises(self):
        from inferno_amd.controller.k8s import ConflictError

        def handler(req):
            return httpx.Response(409, json={})

        kube = make_kube(handler)
        va = api.VariantAutoscaling(name="v1", namespace="ns", resourceVersion="42")
        import pytest

        with pytest.raises(ConflictError):
            kube.update_va_status(va)

    def test_get_variantautoscaling(self):
        def handler(req):
            assert req.url.path == "/apis/llmd.ai/v1alpha1/namespaces/ns/variantautoscalings/v1"
            return httpx.Response(200, json={
                "metadata": {"name": "v1", "namespace": "ns", "resourceVersion": "7"},
                "spec": {"modelID": "m"}})

        va = make_kube(handler).get_variantautoscaling("ns", "v1")
        assert va.name == "v1" and va.resourceVersion == "7"

    def test_inmemory_conflict_and_rv_bump(self):
        from inferno_amd.controller.k8s import ConflictError, InMemoryKube

        kube = InMemoryKube()
        va = api.VariantAutoscaling(name="v1", namespace="ns")
        kube.add_va(va)
        fresh = kube.get_variantautoscaling("ns", "v1")
        stale = kube.get_variantautoscaling("ns", "v1")
        kube.update_va_status(fresh)  # bumps stored rv
        import pytest

        with pytest.raises(ConflictError):
            kube.update_va_status(stale)
