"""Out-of-process e2e tier against the kube-apiserver stand-in.

The envtest/Kind-equivalent for this image (no kube binaries exist): a real
HTTP apiserver process (inferno_amd.testing.kubeapi) with CRD validation,
resourceVersion concurrency, status subresource, watches, Leases and
ownerReference GC; the vLLM emulator; a scraping Prometheus stand-in served
over TLS; and the REAL controller process (python -m
inferno_amd.controller.main) driven end-to-end.

Mirrors the reference's tiers:
  * internal/controller/suite_test.go:56-93 (envtest bootstrap + CRD apply)
  * test/e2e/e2e_test.go:341-430 (scale-out under load w/ Prometheus
    cross-check of inferno_desired_replicas), :519 (scale-in at idle),
    :299/:632 (ownerReference GC), e2e_suite_test.go:95-110 (leader lease)
  * the wire-compat acid test: the reference's own sample VA
    (deploy/examples/vllm-emulator/vllme-setup/vllme-variantautoscaling.yaml)
    must apply unchanged against our CRD schema.
"""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import sys
import threading
import time

import httpx
import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CRD = os.path.join(REPO, "deploy", "crd", "llmd.ai_variantautoscalings.yaml")
NS_SYS = "workload-variant-autoscaler-system"


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _spawn(cmd, env=None, match="listening on"):
    """Start a subprocess and wait for its announce line; returns (proc, port)."""
    e = dict(os.environ)
    if env:
        e.update(env)
    proc = subprocess.Popen(
        cmd, stdout=subprocess.PIPE, stderr=subprocess.PIPE, env=e, text=True,
        cwd=REPO,
    )
    port = None
    deadline = time.time() + 30
    while time.time() < deadline:
        line = proc.stdout.readline()
        if match in line:
            port = int(line.strip().rsplit(" ", 1)[-1])
            break
        if proc.poll() is not None:
            break
    if port is None:
        err = proc.stderr.read() if proc.poll() is not None else "(no announce line)"
        proc.kill()
        raise RuntimeError(f"failed to start {cmd}: {err[-2000:]}")
    # drain stdout so the pipe never blocks the child
    threading.Thread(target=lambda: [None for _ in proc.stdout], daemon=True).start()
    threading.Thread(target=lambda: [None for _ in proc.stderr], daemon=True).start()
    return proc, port


@pytest.fixture(scope="module")
def kubeapi():
    proc, port = _spawn(
        [sys.executable, "-m", "inferno_amd.testing.kubeapi", "--port", "0",
         "--crd", CRD, "--token", "metrics-secret"],
        match="kubeapi listening on",
    )
    client = httpx.Client(base_url=f"http://127.0.0.1:{port}", timeout=10.0)
    # readiness
    for _ in range(50):
        try:
            if client.get("/healthz").status_code == 200:
                break
        except httpx.HTTPError:
            time.sleep(0.1)
    yield client, port
    client.close()
    proc.send_signal(signal.SIGTERM)
    proc.wait(timeout=10)


def _apply(client: httpx.Client, doc: dict) -> httpx.Response:
    kind = doc["kind"]
    plural = {
        "ConfigMap": ("api/v1", "configmaps"),
        "Deployment": ("apis/apps/v1", "deployments"),
        "VariantAutoscaling": ("apis/llmd.ai/v1alpha1", "variantautoscalings"),
        "Lease": ("apis/coordination.k8s.io/v1", "leases"),
    }[kind]
    ns = doc.get("metadata", {}).get("namespace", "default")
    return client.post(f"/{plural[0]}/namespaces/{ns}/{plural[1]}", json=doc)


class TestCrdWireCompat:
    def test_repo_example_va_applies(self, kubeapi):
        client, _ = kubeapi
        with open(os.path.join(REPO, "deploy", "examples",
                               "vllme-variantautoscaling.yaml")) as f:
            docs = [d for d in yaml.safe_load_all(f) if d]
        for doc in docs:
            doc["metadata"]["namespace"] = "wire-test"
            r = _apply(client, doc)
            assert r.status_code == 201, r.text

    def test_reference_sample_va_applies_unchanged(self, kubeapi):
        """The acid test: the reference's own sample VA, byte-for-byte."""
        ref = "/root/reference/deploy/examples/vllm-emulator/vllme-setup/vllme-variantautoscaling.yaml"
        if not os.path.exists(ref):
            pytest.skip("reference checkout not present")
        client, _ = kubeapi
        with open(ref) as f:
            docs = [d for d in yaml.safe_load_all(f) if d]
        assert docs, "no docs in reference sample"
        for doc in docs:
            r = _apply(client, doc)
            assert r.status_code == 201, r.text
            got = r.json()
            assert got["spec"] == doc["spec"]  # nothing pruned or mutated

    def test_invalid_va_rejected(self, kubeapi):
        client, _ = kubeapi
        bad = {
            "apiVersion": "llmd.ai/v1alpha1",
            "kind": "VariantAutoscaling",
            "metadata": {"name": "bad-va", "namespace": "wire-test"},
            "spec": {"sloClassRef": {"name": "x", "key": "y"}},  # no modelID
        }
        r = _apply(client, bad)
        assert r.status_code == 422
        assert "modelID" in r.text


class TestApiMachinery:
    """Real-HTTP API-machinery semantics the InMemoryKube fake can't prove."""

    def _mk_va(self, client, name, ns="machinery"):
        doc = {
            "apiVersion": "llmd.ai/v1alpha1",
            "kind": "VariantAutoscaling",
            "metadata": {"name": name, "namespace": ns,
                         "labels": {"inference.optimization/acceleratorName": "MI355X"}},
            "spec": {
                "modelID": "m/x",
                "sloClassRef": {"name": "svc", "key": "premium.yaml"},
                "modelProfile": {"accelerators": [{
                    "acc": "MI355X", "accCount": 1,
                    "perfParms": {"decodeParms": {"alpha": "10", "beta": "0.2"},
                                  "prefillParms": {"gamma": "2", "delta": "0.01"}},
                    "maxBatchSize": 8}]},
            },
        }
        r = _apply(client, doc)
        assert r.status_code == 201, r.text
        return r.json()

    def test_status_subresource_isolation_and_generation(self, kubeapi):
        client, _ = kubeapi
        va = self._mk_va(client, "va-status")
        path = "/apis/llmd.ai/v1alpha1/namespaces/machinery/variantautoscalings/va-status"
        # status patch: only .status changes, generation stays
        r = client.patch(
            f"{path}/status",
            json={"status": {"desiredOptimizedAlloc": {"numReplicas": 3,
                                                       "accelerator": "MI355X"}},
                  "spec": {"modelID": "SNEAKY"}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        assert r.status_code == 200, r.text
        got = client.get(path).json()
        assert got["spec"]["modelID"] == "m/x"  # spec untouched via /status
        assert got["status"]["desiredOptimizedAlloc"]["numReplicas"] == 3
        assert got["metadata"]["generation"] == 1
        assert int(got["metadata"]["resourceVersion"]) > int(
            va["metadata"]["resourceVersion"]
        )
        # spec patch on the main resource bumps generation, keeps status
        r = client.patch(
            path, json={"spec": {"modelID": "m/y"}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        assert r.status_code == 200
        got = client.get(path).json()
        assert got["metadata"]["generation"] == 2
        assert got["status"]["desiredOptimizedAlloc"]["numReplicas"] == 3

    def test_stale_resource_version_conflicts(self, kubeapi):
        client, _ = kubeapi
        va = self._mk_va(client, "va-conflict")
        path = "/apis/llmd.ai/v1alpha1/namespaces/machinery/variantautoscalings/va-conflict"
        stale_rv = va["metadata"]["resourceVersion"]
        # first writer wins
        r1 = client.patch(
            f"{path}/status",
            json={"metadata": {"resourceVersion": stale_rv},
                  "status": {"desiredOptimizedAlloc": {"numReplicas": 1}}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        assert r1.status_code == 200
        # second writer with the same (now stale) rv conflicts
        r2 = client.patch(
            f"{path}/status",
            json={"metadata": {"resourceVersion": stale_rv},
                  "status": {"desiredOptimizedAlloc": {"numReplicas": 9}}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        assert r2.status_code == 409
        assert client.get(path).json()["status"]["desiredOptimizedAlloc"][
            "numReplicas"] == 1

    def test_merge_patch_null_deletes(self, kubeapi):
        client, _ = kubeapi
        r = _apply(client, {
            "apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {"name": "cm-merge", "namespace": "machinery"},
            "data": {"a": "1", "b": "2"},
        })
        assert r.status_code == 201
        path = "/api/v1/namespaces/machinery/configmaps/cm-merge"
        r = client.patch(path, json={"data": {"a": None, "c": "3"}},
                         headers={"Content-Type": "application/merge-patch+json"})
        assert r.status_code == 200
        assert client.get(path).json()["data"] == {"b": "2", "c": "3"}

    def test_watch_stream_and_resume(self, kubeapi):
        client, port = kubeapi
        events = []

        def consume():
            with httpx.Client(base_url=f"http://127.0.0.1:{port}", timeout=15.0) as c:
                with c.stream(
                    "GET", "/apis/llmd.ai/v1alpha1/variantautoscalings",
                    params={"watch": "1", "timeoutSeconds": "5"},
                ) as r:
                    for line in r.iter_lines():
                        if line:
                            events.append(json.loads(line))

        t = threading.Thread(target=consume)
        t.start()
        time.sleep(0.5)
        self._mk_va(client, "va-watch")
        t.join(timeout=10)
        types = [(e["type"], e["object"]["metadata"]["name"]) for e in events]
        assert ("ADDED", "va-watch") in types

    def test_owner_reference_gc(self, kubeapi):
        client, _ = kubeapi
        r = _apply(client, {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": "owner-dep", "namespace": "machinery"},
            "spec": {"replicas": 1},
        })
        dep = r.json()
        va = self._mk_va(client, "owner-dep")  # VA named after its deployment
        path = "/apis/llmd.ai/v1alpha1/namespaces/machinery/variantautoscalings/owner-dep"
        r = client.patch(
            path,
            json={"metadata": {"ownerReferences": [{
                "apiVersion": "apps/v1", "kind": "Deployment",
                "name": "owner-dep", "uid": dep["metadata"]["uid"],
                "controller": True, "blockOwnerDeletion": True}]}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        assert r.status_code == 200
        r = client.delete("/apis/apps/v1/namespaces/machinery/deployments/owner-dep")
        assert r.status_code == 200
        # cascade GC removed the owned VA (ref e2e_test.go:299,632)
        assert client.get(path).status_code == 404

    def test_lease_put_conflict(self, kubeapi):
        client, _ = kubeapi
        r = _apply(client, {
            "apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
            "metadata": {"name": "test-lease", "namespace": "machinery"},
            "spec": {"holderIdentity": "a", "leaseDurationSeconds": 15},
        })
        lease = r.json()
        path = "/apis/coordination.k8s.io/v1/namespaces/machinery/leases/test-lease"
        # renew with the fetched rv: ok
        lease["spec"]["holderIdentity"] = "a2"
        assert client.put(path, json=lease).status_code == 200
        # renew again with the OLD rv: conflict
        lease["spec"]["holderIdentity"] = "b"
        assert client.put(path, json=lease).status_code == 409

    def test_tokenreview(self, kubeapi):
        client, _ = kubeapi
        r = client.post("/apis/authentication.k8s.io/v1/tokenreviews",
                        json={"spec": {"token": "metrics-secret"}})
        assert r.json()["status"]["authenticated"] is True
        r = client.post("/apis/authentication.k8s.io/v1/tokenreviews",
                        json={"spec": {"token": "nope"}})
        assert r.json()["status"]["authenticated"] is False


class TestLeaderElectionE2E:
    def test_single_leader_over_http(self, kubeapi):
        from inferno_amd.controller.k8s import HttpKube
        from inferno_amd.controller.leader import LeaderElector

        _, port = kubeapi
        kube_a = HttpKube(base_url=f"http://127.0.0.1:{port}")
        kube_b = HttpKube(base_url=f"http://127.0.0.1:{port}")
        a = LeaderElector(kube_a, "e2e-lease", "machinery", "A", lease_seconds=2)
        b = LeaderElector(kube_b, "e2e-lease", "machinery", "B", lease_seconds=2)
        got_a = a.try_acquire()
        got_b = b.try_acquire()
        assert got_a is True and got_b is False  # exactly one leader
        # A stops renewing; after expiry B takes over
        time.sleep(2.5)
        assert b.try_acquire() is True


class TestMergePatchProperties:
    """RFC 7386 merge-patch semantics of the apiserver stand-in, checked
    against an independent reference implementation over randomized docs."""

    @staticmethod
    def _rfc7386(target, patch):
        # independent re-implementation straight from the RFC pseudocode
        if not isinstance(patch, dict):
            return patch
        if not isinstance(target, dict):
            target = {}
        result = dict(target)
        for k, v in patch.items():
            if v is None:
                result.pop(k, None)
            else:
                result[k] = TestMergePatchProperties._rfc7386(result.get(k), v)
        return result

    def test_rfc_examples(self):
        from inferno_amd.testing.kubeapi import merge_patch

        cases = [
            ({"a": "b"}, {"a": "c"}, {"a": "c"}),
            ({"a": "b"}, {"b": "c"}, {"a": "b", "b": "c"}),
            ({"a": "b"}, {"a": None}, {}),
            ({"a": "b", "b": "c"}, {"a": None}, {"b": "c"}),
            ({"a": ["b"]}, {"a": "c"}, {"a": "c"}),
            ({"a": "c"}, {"a": ["b"]}, {"a": ["b"]}),
            ({"a": {"b": "c"}}, {"a": {"b": "d", "c": None}}, {"a": {"b": "d"}}),
            ({"a": [{"b": "c"}]}, {"a": [1]}, {"a": [1]}),
            (["a", "b"], ["c", "d"], ["c", "d"]),
            ({"a": "b"}, ["c"], ["c"]),
            ({"a": "foo"}, None, None),
            ({"a": "foo"}, "bar", "bar"),
            ({"e": None}, {"a": 1}, {"e": None, "a": 1}),
            ([1, 2], {"a": "b", "c": None}, {"a": "b"}),
            ({}, {"a": {"bb": {"ccc": None}}}, {"a": {"bb": {}}}),
        ]
        for target, patch, want in cases:
            assert merge_patch(target, patch) == want, (target, patch)

    def test_randomized_against_reference(self):
        import random

        from inferno_amd.testing.kubeapi import merge_patch

        rng = random.Random(99)

        def rand_doc(depth=0):
            r = rng.random()
            if depth > 3 or r < 0.25:
                return rng.choice([None, 1, "x", True, [1, 2], "y"])
            return {
                rng.choice("abcde"): rand_doc(depth + 1)
                for _ in range(rng.randint(0, 4))
            }

        for _ in range(300):
            t, p = rand_doc(), rand_doc()
            assert merge_patch(t, p) == self._rfc7386(t, p)


class TestNodeInventoryOverHttp:
    """Cluster GPU inventory through the real HTTP apiserver stand-in
    (nodes are cluster-scoped — a path the namespaced fakes never hit)."""

    def test_list_nodes_and_collect(self, kubeapi):
        client, port = kubeapi
        for i, labels in enumerate([
            {"amd.com/gpu.count": "4", "amd.com/gpu.product": "MI355X",
             "amd.com/gpu.memory": "288GB"},
            {"nvidia.com/gpu.count": "2", "nvidia.com/gpu.product": "A100",
             "nvidia.com/gpu.memory": "80GB"},
        ]):
            r = client.post("/api/v1/nodes", json={
                "apiVersion": "v1", "kind": "Node",
                "metadata": {"name": f"gpu-node-{i}", "labels": labels}})
            assert r.status_code == 201, r.text

        from inferno_amd.controller import collector
        from inferno_amd.controller.k8s import HttpKube

        kube = HttpKube(base_url=f"http://127.0.0.1:{port}")
        nodes = kube.list_nodes()
        assert {n.name for n in nodes} >= {"gpu-node-0", "gpu-node-1"}
        inv = collector.collect_inventory_k8s(kube)
        assert inv["amd.com"]["MI355X"]["count"] == 4
        assert inv["nvidia.com"]["A100"]["count"] == 2


class TestWatchResume:
    def test_resume_skips_replayed_events(self, kubeapi):
        """A watch opened at the store's current resourceVersion must not
        replay history (the controller's reconnect path relies on this to
        avoid spurious wake-ups)."""
        client, port = kubeapi
        # create one VA to have history, then note the current rv
        TestApiMachinery()._mk_va(client, "va-resume-old")
        lst = client.get("/apis/llmd.ai/v1alpha1/variantautoscalings").json()
        rv = lst["metadata"]["resourceVersion"]

        events = []

        def consume():
            with httpx.Client(base_url=f"http://127.0.0.1:{port}",
                              timeout=15.0) as c:
                with c.stream(
                    "GET", "/apis/llmd.ai/v1alpha1/variantautoscalings",
                    params={"watch": "1", "timeoutSeconds": "4",
                            "resourceVersion": rv},
                ) as r:
                    for line in r.iter_lines():
                        if line:
                            events.append(json.loads(line))

        t = threading.Thread(target=consume)
        t.start()
        time.sleep(0.5)
        TestApiMachinery()._mk_va(client, "va-resume-new")
        t.join(timeout=10)
        names = [e["object"]["metadata"]["name"] for e in events]
        assert "va-resume-new" in names
        assert "va-resume-old" not in names  # history not replayed


class TestScaleSubresource:
    def test_scale_get_and_patch(self, kubeapi):
        """autoscaling/v1 Scale subresource — what a REAL HPA controller
        talks to when scaling the Deployment."""
        client, _ = kubeapi
        r = _apply(client, {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": "scaled-dep", "namespace": "machinery"},
            "spec": {"replicas": 1}})
        assert r.status_code == 201
        path = "/apis/apps/v1/namespaces/machinery/deployments/scaled-dep/scale"
        got = client.get(path).json()
        assert got["kind"] == "Scale" and got["spec"]["replicas"] == 1
        r = client.patch(path, json={"spec": {"replicas": 5}},
                         headers={"Content-Type": "application/merge-patch+json"})
        assert r.status_code == 200 and r.json()["spec"]["replicas"] == 5
        dep = client.get(
            "/apis/apps/v1/namespaces/machinery/deployments/scaled-dep").json()
        assert dep["spec"]["replicas"] == 5
