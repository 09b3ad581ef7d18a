"""HPA stand-in: the external-metrics actuation loop.

Plays the role of HPA + prometheus-adapter in the reference's production
path (docs/integrations/hpa-integration.md; the OpenShift e2e asserts VA
recommendation -> HPA scale-up -> Deployment replica convergence,
test/e2e-openshift/sharegpt_scaleup_test.go:123-214): polls the controller's
/metrics endpoint for the `inferno_desired_replicas` gauge and patches the
target Deployment's spec.replicas (and mirrors it into status.replicas, as
a healthy rollout would converge to).

Run: python -m inferno_amd.testing.hpa --metrics-url URL --kube-url URL \
        --namespace NS [--interval 2]
"""
from __future__ import annotations

import argparse
import re
import time

import httpx

_GAUGE_RE = re.compile(
    r'^inferno_desired_replicas\{([^}]*)\}\s+([0-9.eE+-]+)\s*$', re.M
)
_LABEL_RE = re.compile(r'(\w+)="([^"]*)"')


def read_desired(metrics_text: str) -> dict[tuple[str, str], int]:
    """Parse {(namespace, variant_name): desired} from an exposition dump."""
    out: dict[tuple[str, str], int] = {}
    for m in _GAUGE_RE.finditer(metrics_text):
        labels = dict(_LABEL_RE.findall(m.group(1)))
        name = labels.get("variant_name", "")
        ns = labels.get("namespace", "")
        if name:
            out[(ns, name)] = int(float(m.group(2)))
    return out


def reconcile_once(metrics_url: str, kube: httpx.Client,
                   verify=None) -> list[tuple[str, str, int]]:
    """One HPA pass; returns the (ns, name, replicas) patches applied."""
    r = httpx.get(metrics_url, timeout=10, verify=verify if verify is not None
                  else True)
    r.raise_for_status()
    applied = []
    for (ns, name), desired in read_desired(r.text).items():
        if desired < 1:
            continue  # HPA floors at minReplicas >= 1
        path = f"/apis/apps/v1/namespaces/{ns}/deployments/{name}"
        cur = kube.get(path)
        if cur.status_code != 200:
            continue
        spec_replicas = (cur.json().get("spec") or {}).get("replicas", 0)
        if spec_replicas == desired:
            continue
        pr = kube.patch(
            path, json={"spec": {"replicas": desired}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        if pr.status_code != 200:
            continue
        # mirror into status.replicas (the kubelet/rollout convergence a
        # healthy Deployment would reach) — status is a SUBRESOURCE, the
        # main-resource patch rightly cannot touch it
        kube.patch(
            f"{path}/status", json={"status": {"replicas": desired}},
            headers={"Content-Type": "application/merge-patch+json"},
        )
        applied.append((ns, name, desired))
    return applied


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--metrics-url", required=True)
    p.add_argument("--kube-url", required=True)
    p.add_argument("--interval", type=float, default=2.0)
    args = p.parse_args()
    kube = httpx.Client(base_url=args.kube_url, timeout=10)
    print("hpa-standin running", flush=True)
    while True:
        try:
            for ns, name, n in reconcile_once(args.metrics_url, kube):
                print(f"hpa: scaled {ns}/{name} -> {n}", flush=True)
        except Exception as e:  # noqa: BLE001
            print(f"hpa: pass failed: {e}", flush=True)
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
