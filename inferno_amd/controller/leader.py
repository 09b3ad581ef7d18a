"""Lease-based leader election (coordination.k8s.io/v1), the analogue of
controller-runtime's leader election used by cmd/main.go:201-219 with
LeaderElectionID "72dd1cf1.llm-d.ai".

Fails CLOSED, matching controller-runtime: on any Lease API error,
unexpected status, or unreachable coordination API the instance is NOT the
leader and retries with backoff. The only fail-open path is the explicit
in-process fake (no HTTP client at all — unit-test double) or running with
``--no-leader-elect``, in which case no elector is constructed.
"""
from __future__ import annotations

import time
from datetime import datetime, timezone


def _now() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%f0Z")


class LeaderElector:
    def __init__(self, kube, lease_name: str, namespace: str, identity: str,
                 lease_seconds: int = 15):
        self.kube = kube
        self.lease_name = lease_name
        self.namespace = namespace
        self.identity = identity
        self.lease_seconds = lease_seconds
        self._is_leader = False
        self._last_renew = 0.0

    def _lease_path(self) -> str:
        return (
            f"/apis/coordination.k8s.io/v1/namespaces/{self.namespace}"
            f"/leases/{self.lease_name}"
        )

    def try_acquire(self) -> bool:
        """Acquire or renew the lease; returns True while we are the leader.

        Error behavior is fail-closed: any API error or exception drops
        leadership immediately (ref cmd/main.go:201-219 — controller-runtime
        cancels the manager context when renewal fails)."""
        client = getattr(self.kube, "_client", None)
        if client is None:
            return True  # in-memory fake: no election (test double only)
        now = time.time()
        if self._is_leader and now - self._last_renew < self.lease_seconds / 3:
            return True
        try:
            r = client.get(self._lease_path())
            if r.status_code == 404:
                body = {
                    "apiVersion": "coordination.k8s.io/v1",
                    "kind": "Lease",
                    "metadata": {"name": self.lease_name, "namespace": self.namespace},
                    "spec": {
                        "holderIdentity": self.identity,
                        "leaseDurationSeconds": self.lease_seconds,
                        "acquireTime": _now(),
                        "renewTime": _now(),
                    },
                }
                cr = client.post(
                    f"/apis/coordination.k8s.io/v1/namespaces/{self.namespace}/leases",
                    json=body,
                )
                # 409 = another replica created it first: not the leader
                self._is_leader = cr.status_code in (200, 201)
            elif r.status_code == 200:
                lease = r.json()
                spec = lease.get("spec", {})
                holder = spec.get("holderIdentity", "")
                renew = spec.get("renewTime", "")
                expired = True
                if renew:
                    try:
                        rt = datetime.strptime(renew[:19], "%Y-%m-%dT%H:%M:%S").replace(
                            tzinfo=timezone.utc
                        )
                        expired = (
                            datetime.now(timezone.utc) - rt
                        ).total_seconds() > spec.get("leaseDurationSeconds", 15)
                    except ValueError:
                        pass
                if holder == self.identity or expired or not holder:
                    lease["spec"]["holderIdentity"] = self.identity
                    lease["spec"]["renewTime"] = _now()
                    # PUT carries the fetched resourceVersion: a concurrent
                    # takeover surfaces as 409 and we lose leadership
                    ur = client.put(self._lease_path(), json=lease)
                    self._is_leader = ur.status_code == 200
                else:
                    self._is_leader = False
            else:
                self._is_leader = False  # API error: fail closed
        except Exception:
            self._is_leader = False  # coordination API unreachable: fail closed
        self._last_renew = now
        return self._is_leader
