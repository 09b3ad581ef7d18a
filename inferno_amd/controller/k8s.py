"""Minimal Kubernetes client layer.

``HttpKube`` talks to the API server over REST (httpx, in-cluster service
account or kubeconfig-provided token/CA) for the resources the controller
needs: VariantAutoscalings (CRD), Deployments, ConfigMaps, Leases (leader
election). ``InMemoryKube`` is the envtest-style fake used by the test
suites (SURVEY.md section 4 tier 2/3).
"""
from __future__ import annotations

import copy
import os
from dataclasses import dataclass, field
from typing import Any, Optional, Protocol

from ..api import v1alpha1 as api


class ConflictError(Exception):
    """Optimistic-concurrency conflict (HTTP 409): the object's
    resourceVersion moved under us. Mirrors the conflict the reference's
    ``Status().Update()`` gets from apimachinery (internal/utils/utils.go:91-104
    retries it with backoff)."""


@dataclass
class Deployment:
    name: str
    namespace: str
    replicas: int = 1  # spec.replicas
    status_replicas: int = 0  # status.replicas
    uid: str = ""
    labels: dict[str, str] = field(default_factory=dict)


class KubeClient(Protocol):  # pragma: no cover - protocol
    def list_variantautoscalings(self) -> list[api.VariantAutoscaling]: ...

    def get_configmap(self, namespace: str, name: str) -> Optional[dict[str, str]]: ...

    def get_deployment(self, namespace: str, name: str) -> Optional[Deployment]: ...

    def update_va_status(self, va: api.VariantAutoscaling) -> None: ...

    def set_owner_reference(self, va: api.VariantAutoscaling, deploy: Deployment) -> None: ...


@dataclass
class Node:
    """Cluster node as the inventory collector sees it: name + labels
    (the reference's emulated-GPU convention labels nodes with
    ``{vendor}.com/gpu.count|.product|.memory``, deploy/kind-emulator/
    setup.sh:120-133)."""

    name: str
    labels: dict[str, str] = field(default_factory=dict)


class InMemoryKube:
    """In-memory fake of the API server (test double)."""

    def __init__(self) -> None:
        self.vas: dict[tuple[str, str], api.VariantAutoscaling] = {}
        self.configmaps: dict[tuple[str, str], dict[str, str]] = {}
        self.deployments: dict[tuple[str, str], Deployment] = {}
        self.nodes: dict[str, Node] = {}
        self.status_updates: list[dict[str, Any]] = []
        self._rv_counter = 0

    def _next_rv(self) -> str:
        self._rv_counter += 1
        return str(self._rv_counter)

    # -- setup helpers -------------------------------------------------
    def add_va(self, va: api.VariantAutoscaling) -> None:
        if not va.resourceVersion:
            va.resourceVersion = self._next_rv()
        self.vas[(va.namespace, va.name)] = va

    def add_configmap(self, namespace: str, name: str, data: dict[str, str]) -> None:
        self.configmaps[(namespace, name)] = dict(data)

    def add_deployment(self, deploy: Deployment) -> None:
        self.deployments[(deploy.namespace, deploy.name)] = deploy

    def add_node(self, node: Node) -> None:
        self.nodes[node.name] = node

    def list_nodes(self) -> list[Node]:
        return [copy.deepcopy(n) for n in self.nodes.values()]

    # -- KubeClient ----------------------------------------------------
    def list_variantautoscalings(self) -> list[api.VariantAutoscaling]:
        return [copy.deepcopy(v) for v in self.vas.values()]

    def get_variantautoscaling(self, namespace: str, name: str) -> Optional[api.VariantAutoscaling]:
        va = self.vas.get((namespace, name))
        return copy.deepcopy(va) if va is not None else None

    def get_configmap(self, namespace: str, name: str) -> Optional[dict[str, str]]:
        cm = self.configmaps.get((namespace, name))
        return dict(cm) if cm is not None else None

    def get_deployment(self, namespace: str, name: str) -> Optional[Deployment]:
        return self.deployments.get((namespace, name))

    def update_va_status(self, va: api.VariantAutoscaling) -> None:
        key = (va.namespace, va.name)
        if key not in self.vas:
            raise KeyError(f"VariantAutoscaling {key} not found")
        stored = self.vas[key]
        # optimistic concurrency: a stale resourceVersion conflicts instead
        # of clobbering (apiserver PUT/patch-with-rv semantics)
        if va.resourceVersion and stored.resourceVersion and (
            va.resourceVersion != stored.resourceVersion
        ):
            raise ConflictError(
                f"resourceVersion conflict on {key}: "
                f"have {va.resourceVersion}, stored {stored.resourceVersion}"
            )
        stored.status = copy.deepcopy(va.status)
        stored.resourceVersion = self._next_rv()
        va.resourceVersion = stored.resourceVersion
        self.status_updates.append(api.va_to_json(va)["status"])

    def set_owner_reference(self, va: api.VariantAutoscaling, deploy: Deployment) -> None:
        ref = {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "name": deploy.name,
            "uid": deploy.uid,
            "controller": True,
            "blockOwnerDeletion": True,
        }
        key = (va.namespace, va.name)
        stored = self.vas.get(key)
        refs = [r for r in (stored.ownerReferences if stored else []) if r.get("uid") != deploy.uid]
        refs.append(ref)
        if stored is not None:
            stored.ownerReferences = refs
        va.ownerReferences = refs


DEFAULT_SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class HttpKube:
    """REST client against the API server (in-cluster by default)."""

    def __init__(
        self,
        base_url: Optional[str] = None,
        token: Optional[str] = None,
        ca_cert: Optional[str] = None,
        verify: bool | str = True,
        timeout: float = 15.0,
    ):
        import httpx

        if base_url is None:
            # KUBE_API_URL overrides the in-cluster default (used by the
            # envtest-equivalent e2e tier to point at the apiserver stand-in)
            base_url = os.environ.get("KUBE_API_URL") or None
        if base_url is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            base_url = f"https://{host}:{port}"
        if token is None:
            token_path = os.path.join(DEFAULT_SA_DIR, "token")
            if os.path.exists(token_path):
                with open(token_path) as f:
                    token = f.read().strip()
        if ca_cert is None:
            ca_path = os.path.join(DEFAULT_SA_DIR, "ca.crt")
            if os.path.exists(ca_path):
                ca_cert = ca_path
        headers = {"Content-Type": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._client = httpx.Client(
            base_url=base_url,
            headers=headers,
            verify=ca_cert if ca_cert else verify,
            timeout=timeout,
        )

    # -- helpers -------------------------------------------------------
    def _get(self, path: str) -> Optional[dict]:
        r = self._client.get(path)
        if r.status_code == 404:
            return None
        r.raise_for_status()
        return r.json()

    # -- KubeClient ----------------------------------------------------
    def list_variantautoscalings(self) -> list[api.VariantAutoscaling]:
        doc = self._get(f"/apis/{api.GROUP}/{api.VERSION}/{api.PLURAL}")
        if doc is None:
            return []
        return [api.va_from_json(item) for item in doc.get("items", [])]

    def get_variantautoscaling(self, namespace: str, name: str) -> Optional[api.VariantAutoscaling]:
        doc = self._get(
            f"/apis/{api.GROUP}/{api.VERSION}/namespaces/{namespace}/{api.PLURAL}/{name}"
        )
        return None if doc is None else api.va_from_json(doc)

    def get_configmap(self, namespace: str, name: str) -> Optional[dict[str, str]]:
        doc = self._get(f"/api/v1/namespaces/{namespace}/configmaps/{name}")
        return None if doc is None else dict(doc.get("data", {}) or {})

    def list_nodes(self) -> list[Node]:
        doc = self._get("/api/v1/nodes")
        if doc is None:
            return []
        out = []
        for item in doc.get("items", []) or []:
            meta = item.get("metadata", {}) or {}
            out.append(Node(name=meta.get("name", ""),
                            labels=dict(meta.get("labels", {}) or {})))
        return out

    def get_deployment(self, namespace: str, name: str) -> Optional[Deployment]:
        doc = self._get(f"/apis/apps/v1/namespaces/{namespace}/deployments/{name}")
        if doc is None:
            return None
        meta = doc.get("metadata", {}) or {}
        spec = doc.get("spec", {}) or {}
        status = doc.get("status", {}) or {}
        return Deployment(
            name=meta.get("name", name),
            namespace=meta.get("namespace", namespace),
            replicas=int(spec.get("replicas", 1) or 0),
            status_replicas=int(status.get("replicas", 0) or 0),
            uid=meta.get("uid", ""),
            labels=dict(meta.get("labels", {}) or {}),
        )

    def update_va_status(self, va: api.VariantAutoscaling) -> None:
        path = (
            f"/apis/{api.GROUP}/{api.VERSION}/namespaces/{va.namespace}/"
            f"{api.PLURAL}/{va.name}/status"
        )
        meta: dict = {"name": va.name, "namespace": va.namespace}
        # carry resourceVersion so concurrent writers conflict (409) instead
        # of last-write-wins, matching the reference's Status().Update()
        if va.resourceVersion:
            meta["resourceVersion"] = va.resourceVersion
        body = {
            "apiVersion": api.API_VERSION,
            "kind": api.KIND,
            "metadata": meta,
            "status": api.va_to_json(va)["status"],
        }
        r = self._client.patch(
            path, json=body, headers={"Content-Type": "application/merge-patch+json"}
        )
        if r.status_code == 409:
            raise ConflictError(f"resourceVersion conflict updating {va.namespace}/{va.name}")
        r.raise_for_status()
        try:
            new_rv = ((r.json().get("metadata") or {}).get("resourceVersion"))
            if new_rv:
                va.resourceVersion = str(new_rv)
        except ValueError:
            pass

    def watch_events(self, resource_version: str = "", timeout_seconds: int = 60):
        """Stream watch events for VariantAutoscalings (the reference's
        controller-runtime For(VA) watch with a create-only event filter,
        controller.go:456-487). Yields (event_type, VariantAutoscaling).
        """
        params = {"watch": "1", "timeoutSeconds": str(timeout_seconds)}
        if resource_version:
            params["resourceVersion"] = resource_version
        import json as _json

        with self._client.stream(
            "GET", f"/apis/{api.GROUP}/{api.VERSION}/{api.PLURAL}", params=params,
            timeout=timeout_seconds + 10,
        ) as r:
            r.raise_for_status()
            for line in r.iter_lines():
                if not line:
                    continue
                try:
                    evt = _json.loads(line)
                except ValueError:
                    continue
                etype = evt.get("type", "")
                obj = evt.get("object", {}) or {}
                if obj.get("kind") != api.KIND:
                    continue
                yield etype, api.va_from_json(obj)

    def watch_configmap_events(self, namespace: str, names: set[str],
                               timeout_seconds: int = 60,
                               resource_version: str = ""):
        """Stream watch events for the controller's ConfigMaps (the
        reference's Watches(ConfigMap) registration, controller.go:456-487).
        Yields (event_type, name, resourceVersion) — callers thread the last
        resourceVersion back in on reconnect so history is not replayed."""
        import json as _json

        params = {"watch": "1", "timeoutSeconds": str(timeout_seconds)}
        if resource_version:
            params["resourceVersion"] = resource_version
        with self._client.stream(
            "GET", f"/api/v1/namespaces/{namespace}/configmaps", params=params,
            timeout=timeout_seconds + 10,
        ) as r:
            r.raise_for_status()
            for line in r.iter_lines():
                if not line:
                    continue
                try:
                    evt = _json.loads(line)
                except ValueError:
                    continue
                meta = (evt.get("object") or {}).get("metadata") or {}
                name = meta.get("name", "")
                rv = str(meta.get("resourceVersion", "") or "")
                if name in names:
                    yield evt.get("type", ""), name, rv

    def set_owner_reference(self, va: api.VariantAutoscaling, deploy: Deployment) -> None:
        path = f"/apis/{api.GROUP}/{api.VERSION}/namespaces/{va.namespace}/{api.PLURAL}/{va.name}"
        ref = {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "name": deploy.name,
            "uid": deploy.uid,
            "controller": True,
            "blockOwnerDeletion": True,
        }
        refs = [r for r in va.ownerReferences if r.get("uid") != deploy.uid]
        refs.append(ref)
        body = {"metadata": {"ownerReferences": refs}}
        r = self._client.patch(
            path, json=body, headers={"Content-Type": "application/merge-patch+json"}
        )
        r.raise_for_status()
        va.ownerReferences = refs
