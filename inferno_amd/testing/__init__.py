"""Test infrastructure that stands in for cluster services.

``kubeapi``  — wire-faithful Kubernetes API server stand-in (real HTTP,
              CRD schema validation, resourceVersion optimistic concurrency,
              status subresource, watch streams, Lease API, merge-patch,
              ownerReference GC). The envtest-equivalent tier
              (SURVEY.md section 4 tier 2; ref internal/controller/
              suite_test.go:56-93) for an image with no kube binaries.
``promstub`` — minimal Prometheus: scrapes real /metrics endpoints and
              answers the exact PromQL shapes the collector issues
              (sum(rate(m{..}[1m])) and ratios) over /api/v1/query.
"""
