"""Kubernetes API server stand-in (envtest-equivalent tier).

A real HTTP process implementing the API-machinery semantics the controller
depends on, so the controller binary can be tested end-to-end out of process
without kube-apiserver/etcd binaries (none exist in this image):

* CRD registration from YAML with openAPIV3Schema validation (subset:
  type/required/properties/items/minLength/minItems/enum) — the reference's
  sample VA must apply unchanged (wire-compat acid test; ref
  deploy/examples/vllm-emulator/vllme-setup/vllme-variantautoscaling.yaml).
* resourceVersion optimistic concurrency: PUT/merge-patch carrying a stale
  metadata.resourceVersion gets 409 Conflict.
* Status subresource isolation: PATCH on /status touches only .status (and
  bumps resourceVersion, not generation); spec edits bump generation.
* RFC 7386 merge-patch.
* Watch streams (?watch=1&timeoutSeconds=N) with typed ADDED/MODIFIED/DELETED
  events, resumable from a resourceVersion.
* coordination.k8s.io/v1 Leases (leader election) with conflict-on-PUT.
* authentication.k8s.io/v1 TokenReview (static token set).
* ownerReference cascade GC on owner deletion.

Run: python -m inferno_amd.testing.kubeapi --port 0 [--crd path.yaml] \
        [--token t1,t2]
Prints "kubeapi listening on <port>" on stdout when ready.

Mirrors the role of envtest in the reference's tiers 2/4 (SURVEY.md section 4;
internal/controller/suite_test.go:56-93, test/e2e/e2e_test.go).
"""
from __future__ import annotations

import argparse
import asyncio
import copy
import json
import re
import threading
import time
import uuid
from typing import Any, Optional

import yaml


# ---------------------------------------------------------------------------
# RFC 7386 merge patch
# ---------------------------------------------------------------------------
def merge_patch(target: Any, patch: Any) -> Any:
    if not isinstance(patch, dict):
        return copy.deepcopy(patch)
    if not isinstance(target, dict):
        target = {}
    result = dict(target)
    for k, v in patch.items():
        if v is None:
            result.pop(k, None)
        else:
            result[k] = merge_patch(result.get(k), v)
    return result


# ---------------------------------------------------------------------------
# openAPIV3Schema validation (the subset CRDs here use)
# ---------------------------------------------------------------------------
class ValidationError(Exception):
    pass


def validate_schema(obj: Any, schema: dict, path: str = "") -> None:
    t = schema.get("type")
    if t == "object":
        if not isinstance(obj, dict):
            raise ValidationError(f"{path or '.'}: expected object, got {type(obj).__name__}")
        for req in schema.get("required", []):
            if req not in obj:
                raise ValidationError(f"{path}.{req}: required field missing")
        props = schema.get("properties", {})
        for k, v in obj.items():
            if k in props:
                validate_schema(v, props[k], f"{path}.{k}")
            # unknown fields are pruned-tolerant (structural schema default)
    elif t == "array":
        if not isinstance(obj, list):
            raise ValidationError(f"{path}: expected array")
        if "minItems" in schema and len(obj) < schema["minItems"]:
            raise ValidationError(f"{path}: fewer than minItems={schema['minItems']}")
        item_schema = schema.get("items")
        if item_schema:
            for i, it in enumerate(obj):
                validate_schema(it, item_schema, f"{path}[{i}]")
    elif t == "string":
        if not isinstance(obj, str):
            raise ValidationError(f"{path}: expected string, got {type(obj).__name__}")
        if "minLength" in schema and len(obj) < schema["minLength"]:
            raise ValidationError(f"{path}: shorter than minLength={schema['minLength']}")
        if "enum" in schema and obj not in schema["enum"]:
            raise ValidationError(f"{path}: {obj!r} not in enum {schema['enum']}")
    elif t == "integer":
        if isinstance(obj, bool) or not isinstance(obj, int):
            raise ValidationError(f"{path}: expected integer, got {type(obj).__name__}")
        if "minimum" in schema and obj < schema["minimum"]:
            raise ValidationError(f"{path}: below minimum={schema['minimum']}")
    elif t == "number":
        if isinstance(obj, bool) or not isinstance(obj, (int, float)):
            raise ValidationError(f"{path}: expected number")
    elif t == "boolean":
        if not isinstance(obj, bool):
            raise ValidationError(f"{path}: expected boolean")
    # no declared type: accept anything (x-kubernetes-preserve-unknown-fields)


# ---------------------------------------------------------------------------
# object store with events
# ---------------------------------------------------------------------------
class ResourceKind:
    def __init__(self, group: str, version: str, plural: str, kind: str,
                 namespaced: bool = True, schema: Optional[dict] = None,
                 has_status: bool = False):
        self.group = group
        self.version = version
        self.plural = plural
        self.kind = kind
        self.namespaced = namespaced
        self.schema = schema
        self.has_status = has_status

    @property
    def api_version(self) -> str:
        return f"{self.group}/{self.version}" if self.group else self.version


BUILTIN_KINDS = [
    ResourceKind("", "v1", "configmaps", "ConfigMap"),
    ResourceKind("", "v1", "nodes", "Node", namespaced=False, has_status=True),
    ResourceKind("apps", "v1", "deployments", "Deployment", has_status=True),
    ResourceKind("coordination.k8s.io", "v1", "leases", "Lease"),
]


class Store:
    """Versioned object store + event log (the etcd stand-in)."""

    def __init__(self):
        self._lock = threading.Lock()
        self._rv = 0
        self.kinds: dict[str, ResourceKind] = {}  # plural -> kind
        self.objects: dict[tuple[str, str, str], dict] = {}  # (plural, ns, name)
        self.events: list[tuple[int, str, str, dict]] = []  # (rv, type, plural, obj)
        for k in BUILTIN_KINDS:
            self.kinds[k.plural] = k

    def next_rv(self) -> int:
        self._rv += 1
        return self._rv

    def register_crd(self, crd_doc: dict) -> None:
        spec = crd_doc["spec"]
        names = spec["names"]
        version = next(v for v in spec["versions"] if v.get("served", True))
        schema = (version.get("schema") or {}).get("openAPIV3Schema")
        self.kinds[names["plural"]] = ResourceKind(
            group=spec["group"],
            version=version["name"],
            plural=names["plural"],
            kind=names["kind"],
            namespaced=spec.get("scope", "Namespaced") == "Namespaced",
            schema=schema,
            has_status="status" in (version.get("subresources") or {}),
        )

    # cap on retained watch history; resuming from an rv older than the
    # window behaves like the real apiserver's compaction (events are simply
    # gone — clients re-list). Bounds memory for long-lived `stack` runs.
    MAX_EVENTS = 50_000

    # -- CRUD ----------------------------------------------------------
    def _emit(self, etype: str, plural: str, obj: dict) -> None:
        self.events.append((int(obj["metadata"]["resourceVersion"]), etype, plural,
                            copy.deepcopy(obj)))
        if len(self.events) > self.MAX_EVENTS:
            del self.events[: len(self.events) - self.MAX_EVENTS]

    def create(self, plural: str, ns: str, body: dict) -> dict:
        kind = self.kinds[plural]
        with self._lock:
            meta = body.setdefault("metadata", {})
            name = meta.get("name", "")
            if not name:
                raise ValidationError("metadata.name required")
            if kind.namespaced:
                meta["namespace"] = ns
            key = (plural, ns if kind.namespaced else "", name)
            if key in self.objects:
                raise KeyError("exists")
            if kind.schema:
                validate_schema(body, kind.schema)
            meta["uid"] = meta.get("uid") or str(uuid.uuid4())
            meta["resourceVersion"] = str(self.next_rv())
            meta["generation"] = 1
            meta.setdefault(
                "creationTimestamp",
                time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            )
            body["apiVersion"] = kind.api_version
            body["kind"] = kind.kind
            self.objects[key] = body
            self._emit("ADDED", plural, body)
            return copy.deepcopy(body)

    def get(self, plural: str, ns: str, name: str) -> Optional[dict]:
        kind = self.kinds[plural]
        key = (plural, ns if kind.namespaced else "", name)
        with self._lock:
            obj = self.objects.get(key)
            return copy.deepcopy(obj) if obj else None

    def list(self, plural: str, ns: Optional[str] = None) -> list[dict]:
        with self._lock:
            out = []
            for (p, ons, _), obj in self.objects.items():
                if p != plural:
                    continue
                if ns is not None and ons != ns:
                    continue
                out.append(copy.deepcopy(obj))
            return out

    def update(self, plural: str, ns: str, name: str, body: dict,
               subresource: Optional[str] = None) -> dict:
        """PUT semantics: optimistic concurrency on metadata.resourceVersion."""
        kind = self.kinds[plural]
        key = (plural, ns if kind.namespaced else "", name)
        with self._lock:
            cur = self.objects.get(key)
            if cur is None:
                raise KeyError("not found")
            body_rv = str((body.get("metadata") or {}).get("resourceVersion", "") or "")
            if body_rv and body_rv != cur["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"resourceVersion {body_rv} != {cur['metadata']['resourceVersion']}"
                )
            new = self._apply_update(kind, cur, body, subresource)
            self.objects[key] = new
            self._emit("MODIFIED", plural, new)
            return copy.deepcopy(new)

    def patch(self, plural: str, ns: str, name: str, patch: dict,
              subresource: Optional[str] = None) -> dict:
        kind = self.kinds[plural]
        key = (plural, ns if kind.namespaced else "", name)
        with self._lock:
            cur = self.objects.get(key)
            if cur is None:
                raise KeyError("not found")
            patch_rv = str((patch.get("metadata") or {}).get("resourceVersion", "") or "")
            if patch_rv and patch_rv != cur["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"resourceVersion {patch_rv} != {cur['metadata']['resourceVersion']}"
                )
            merged = merge_patch(cur, patch)
            new = self._apply_update(kind, cur, merged, subresource)
            self.objects[key] = new
            self._emit("MODIFIED", plural, new)
            return copy.deepcopy(new)

    def _apply_update(self, kind: ResourceKind, cur: dict, desired: dict,
                      subresource: Optional[str]) -> dict:
        new = copy.deepcopy(cur)
        if subresource == "status":
            # status subresource: ONLY .status changes (generation untouched)
            new["status"] = copy.deepcopy(desired.get("status", {}))
            if kind.schema:
                validate_schema(new, kind.schema)
        else:
            old_spec = cur.get("spec")
            new = copy.deepcopy(desired)
            if kind.has_status:
                # main-resource writes never touch status
                new["status"] = copy.deepcopy(cur.get("status", {}))
            new["metadata"]["uid"] = cur["metadata"]["uid"]
            new["metadata"]["creationTimestamp"] = cur["metadata"]["creationTimestamp"]
            if kind.schema:
                validate_schema(new, kind.schema)
            gen = int(cur["metadata"].get("generation", 1))
            if new.get("spec") != old_spec:
                gen += 1
            new["metadata"]["generation"] = gen
        # immutable identity
        new.setdefault("metadata", {})
        new["metadata"]["name"] = cur["metadata"]["name"]
        if kind.namespaced:
            new["metadata"]["namespace"] = cur["metadata"]["namespace"]
        new["metadata"]["uid"] = cur["metadata"]["uid"]
        new["metadata"]["generation"] = new["metadata"].get(
            "generation", cur["metadata"].get("generation", 1)
        )
        new["metadata"]["creationTimestamp"] = cur["metadata"]["creationTimestamp"]
        new["metadata"]["resourceVersion"] = str(self.next_rv())
        new["apiVersion"] = kind.api_version
        new["kind"] = kind.kind
        return new

    def delete(self, plural: str, ns: str, name: str) -> dict:
        kind = self.kinds[plural]
        key = (plural, ns if kind.namespaced else "", name)
        with self._lock:
            obj = self.objects.pop(key, None)
            if obj is None:
                raise KeyError("not found")
            obj["metadata"]["resourceVersion"] = str(self.next_rv())
            self._emit("DELETED", plural, obj)
            uid = obj["metadata"]["uid"]
        self._gc(uid)
        return obj

    def _gc(self, owner_uid: str) -> None:
        """ownerReference cascade GC (ref e2e ownerRef tests,
        test/e2e/e2e_test.go:299,632)."""
        victims = []
        with self._lock:
            for (plural, ns, name), obj in list(self.objects.items()):
                refs = obj.get("metadata", {}).get("ownerReferences", []) or []
                if any(r.get("uid") == owner_uid for r in refs):
                    victims.append((plural, ns, name))
        for plural, ns, name in victims:
            try:
                self.delete(plural, ns, name)
            except KeyError:
                pass

    def events_since(self, plural: str, rv: int) -> list[tuple[int, str, dict]]:
        with self._lock:
            return [
                (erv, etype, copy.deepcopy(obj))
                for erv, etype, p, obj in self.events
                if p == plural and erv > rv
            ]


class ConflictError(Exception):
    pass


# ---------------------------------------------------------------------------
# HTTP layer (FastAPI)
# ---------------------------------------------------------------------------
def build_app(store: Store, valid_tokens: Optional[set[str]] = None):
    # module-level so FastAPI can resolve the deferred "Request" annotations
    global FastAPI, Request, JSONResponse, StreamingResponse
    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse, StreamingResponse

    app = FastAPI(title="kubeapi-standin")
    app.state.store = store
    valid_tokens = valid_tokens or set()

    def status_err(code: int, reason: str, message: str) -> JSONResponse:
        return JSONResponse(
            status_code=code,
            content={"kind": "Status", "apiVersion": "v1", "status": "Failure",
                     "reason": reason, "message": message, "code": code},
        )

    async def watch_stream(plural: str, start_rv: int, timeout_s: int):
        deadline = time.monotonic() + timeout_s
        rv = start_rv
        while time.monotonic() < deadline:
            for erv, etype, obj in store.events_since(plural, rv):
                rv = max(rv, erv)
                yield json.dumps({"type": etype, "object": obj}) + "\n"
            await asyncio.sleep(0.1)

    def list_or_watch(request: Request, plural: str, ns: Optional[str]):
        kind = store.kinds[plural]
        if request.query_params.get("watch") in ("1", "true"):
            timeout_s = int(request.query_params.get("timeoutSeconds", "60"))
            start_rv = int(request.query_params.get("resourceVersion", "0") or 0)
            return StreamingResponse(
                watch_stream(plural, start_rv, timeout_s),
                media_type="application/json",
            )
        items = store.list(plural, ns)
        return JSONResponse({
            "apiVersion": kind.api_version,
            "kind": kind.kind + "List",
            "metadata": {"resourceVersion": str(store._rv)},
            "items": items,
        })

    async def handle(request: Request, plural: str, ns: Optional[str],
                     name: Optional[str], subresource: Optional[str] = None):
        if plural not in store.kinds:
            return status_err(404, "NotFound", f"unknown resource {plural}")
        method = request.method
        try:
            if method == "GET" and name is None:
                return list_or_watch(request, plural, ns)
            if method == "GET":
                obj = store.get(plural, ns or "", name)
                if obj is None:
                    return status_err(404, "NotFound", f"{plural}/{name} not found")
                return JSONResponse(obj)
            body = json.loads(await request.body() or b"{}")
            if method == "POST":
                return JSONResponse(store.create(plural, ns or "", body), status_code=201)
            if method == "PUT":
                return JSONResponse(store.update(plural, ns or "", name, body, subresource))
            if method == "PATCH":
                ctype = request.headers.get("content-type", "")
                if "merge-patch" not in ctype and "strategic-merge-patch" not in ctype:
                    return status_err(415, "UnsupportedMediaType",
                                      f"unsupported patch type {ctype}")
                return JSONResponse(store.patch(plural, ns or "", name, body, subresource))
            if method == "DELETE":
                return JSONResponse(store.delete(plural, ns or "", name))
            return status_err(405, "MethodNotAllowed", method)
        except ConflictError as e:
            return status_err(409, "Conflict", str(e))
        except ValidationError as e:
            return status_err(422, "Invalid", str(e))
        except KeyError as e:
            if "exists" in str(e):
                return status_err(409, "AlreadyExists", f"{plural}/{name} already exists")
            return status_err(404, "NotFound", f"{plural}/{name} not found")

    # core/v1 cluster-scoped (nodes)
    @app.api_route("/api/v1/nodes", methods=["GET", "POST"])
    async def core_nodes(request: Request):
        return await handle(request, "nodes", "", None)

    @app.api_route("/api/v1/nodes/{name}", methods=["GET", "PUT", "PATCH", "DELETE"])
    async def core_node_named(request: Request, name: str):
        return await handle(request, "nodes", "", name)

    # core/v1
    @app.api_route("/api/v1/namespaces/{ns}/{plural}", methods=["GET", "POST"])
    async def core_collection(request: Request, ns: str, plural: str):
        return await handle(request, plural, ns, None)

    @app.api_route("/api/v1/namespaces/{ns}/{plural}/{name}",
                   methods=["GET", "PUT", "PATCH", "DELETE"])
    async def core_named(request: Request, ns: str, plural: str, name: str):
        return await handle(request, plural, ns, name)

    # TokenReview
    @app.post("/apis/authentication.k8s.io/v1/tokenreviews")
    async def tokenreview(request: Request):
        body = json.loads(await request.body() or b"{}")
        token = (body.get("spec") or {}).get("token", "")
        ok = token in valid_tokens
        return JSONResponse({
            "apiVersion": "authentication.k8s.io/v1",
            "kind": "TokenReview",
            "status": {"authenticated": ok,
                       **({"user": {"username": f"token-user"}} if ok else {})},
        }, status_code=201)

    # grouped APIs: cluster-scope list/watch
    @app.api_route("/apis/{group}/{version}/{plural}", methods=["GET"])
    async def group_cluster(request: Request, group: str, version: str, plural: str):
        if plural not in store.kinds or store.kinds[plural].group != group:
            return status_err(404, "NotFound", f"unknown resource {group}/{plural}")
        return list_or_watch(request, plural, None)

    @app.api_route("/apis/{group}/{version}/namespaces/{ns}/{plural}",
                   methods=["GET", "POST"])
    async def group_collection(request: Request, group: str, version: str, ns: str,
                               plural: str):
        return await handle(request, plural, ns, None)

    @app.api_route("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}",
                   methods=["GET", "PUT", "PATCH", "DELETE"])
    async def group_named(request: Request, group: str, version: str, ns: str,
                          plural: str, name: str):
        return await handle(request, plural, ns, name)

    @app.api_route("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}/status",
                   methods=["GET", "PUT", "PATCH"])
    async def group_status(request: Request, group: str, version: str, ns: str,
                           plural: str, name: str):
        return await handle(request, plural, ns, name, subresource="status")

    @app.api_route("/apis/apps/v1/namespaces/{ns}/deployments/{name}/scale",
                   methods=["GET", "PUT", "PATCH"])
    async def deployment_scale(request: Request, ns: str, name: str):
        """The autoscaling/v1 Scale subresource real HPA controllers use."""
        obj = store.get("deployments", ns, name)
        if obj is None:
            return status_err(404, "NotFound", f"deployments/{name} not found")
        if request.method in ("PUT", "PATCH"):
            body = json.loads(await request.body() or b"{}")
            replicas = int(((body.get("spec") or {}).get("replicas", 0)) or 0)
            try:
                obj = store.patch("deployments", ns, name,
                                  {"spec": {"replicas": replicas}})
            except ConflictError as e:
                return status_err(409, "Conflict", str(e))
        return JSONResponse({
            "apiVersion": "autoscaling/v1",
            "kind": "Scale",
            "metadata": {"name": name, "namespace": ns,
                         "resourceVersion": obj["metadata"]["resourceVersion"]},
            "spec": {"replicas": (obj.get("spec") or {}).get("replicas", 0)},
            "status": {"replicas": (obj.get("status") or {}).get("replicas", 0)},
        })

    @app.get("/healthz")
    async def healthz():
        return JSONResponse({"ok": True})

    @app.get("/version")
    async def version():
        return JSONResponse({"major": "1", "minor": "31",
                             "gitVersion": "v1.31.0-standin"})

    return app


# ---------------------------------------------------------------------------
# process entry
# ---------------------------------------------------------------------------
def apply_yaml_file(store: Store, path: str) -> list[dict]:
    """kubectl-apply a multi-doc YAML file into the store (CRDs register,
    everything else creates)."""
    created = []
    with open(path) as f:
        for doc in yaml.safe_load_all(f):
            if not doc:
                continue
            created.append(apply_doc(store, doc))
    return created


def apply_doc(store: Store, doc: dict) -> dict:
    if doc.get("kind") == "CustomResourceDefinition":
        store.register_crd(doc)
        return doc
    plural = _plural_for(store, doc)
    ns = (doc.get("metadata") or {}).get("namespace", "default")
    return store.create(plural, ns, copy.deepcopy(doc))


def _plural_for(store: Store, doc: dict) -> str:
    kind = doc.get("kind", "")
    for plural, rk in store.kinds.items():
        if rk.kind == kind:
            return plural
    raise ValidationError(f"no registered resource for kind {kind!r}")


def main() -> None:
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--crd", action="append", default=[],
                   help="CRD YAML to register at startup")
    p.add_argument("--apply", action="append", default=[],
                   help="YAML manifests to create at startup")
    p.add_argument("--token", default="", help="comma-separated valid bearer tokens")
    args = p.parse_args()

    store = Store()
    for crd in args.crd:
        for doc in yaml.safe_load_all(open(crd)):
            if doc:
                store.register_crd(doc)
    for path in args.apply:
        apply_yaml_file(store, path)
    tokens = {t for t in args.token.split(",") if t}
    app = build_app(store, tokens)

    config = uvicorn.Config(app, host=args.host, port=args.port, log_level="warning")
    server = uvicorn.Server(config)

    import socket

    # bind explicitly so the chosen ephemeral port can be announced
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind((args.host, args.port))
    port = sock.getsockname()[1]
    print(f"kubeapi listening on {port}", flush=True)
    server.run(sockets=[sock])


if __name__ == "__main__":
    main()
