#!/usr/bin/env bash
# Stand up a local kind cluster that exercises the full autoscaling loop
# with NO real GPUs: emulated vLLM servers (inferno_amd.emulator) report
# vllm:* metrics, nodes carry fake MI355X labels, Prometheus scrapes
# everything, and the controller publishes inferno_desired_replicas.
#
# Usage: ./setup.sh [cluster-name]
set -euo pipefail

CLUSTER="${1:-wva-emulated}"
NS=workload-variant-autoscaler-system
ROOT="$(cd "$(dirname "$0")/../.." && pwd)"

command -v kind >/dev/null || { echo "kind not found" >&2; exit 1; }
command -v kubectl >/dev/null || { echo "kubectl not found" >&2; exit 1; }

echo ">> creating kind cluster ${CLUSTER}"
kind get clusters | grep -qx "${CLUSTER}" || kind create cluster --name "${CLUSTER}" --config - <<'EOF'
kind: Cluster
apiVersion: kind.x-k8s.io/v1alpha4
nodes:
  - role: control-plane
  - role: worker
  - role: worker
EOF

echo ">> labeling workers with emulated MI355X accelerators"
for node in $(kubectl get nodes -l '!node-role.kubernetes.io/control-plane' -o name); do
  kubectl label --overwrite "${node}" \
    amd.com/gpu.family=AI \
    amd.com/gpu.device-id=emulated-mi355x \
    inference.optimization/acceleratorName=MI355X
done

echo ">> installing CRD + controller + ConfigMaps"
kubectl apply -f "${ROOT}/deploy/crd/llmd.ai_variantautoscalings.yaml"
kubectl create namespace "${NS}" --dry-run=client -o yaml | kubectl apply -f -
kubectl apply -f "${ROOT}/deploy/configmap-accelerator-unitcost.yaml"
kubectl apply -f "${ROOT}/deploy/configmap-serviceclass.yaml"
kubectl apply -f "${ROOT}/deploy/controller.yaml"

echo ">> deploying the emulated vLLM server + example VariantAutoscaling"
kubectl apply -f "${ROOT}/deploy/emulator.yaml"
kubectl apply -f "${ROOT}/deploy/examples/vllme-variantautoscaling.yaml"

echo ">> (optional) HPA integration acting on inferno_desired_replicas"
kubectl apply -f "${ROOT}/deploy/examples/hpa-integration.yaml" || true

cat <<DONE

Cluster '${CLUSTER}' is up.

Next steps:
  # install a Prometheus that scrapes the emulator + controller, e.g.:
  #   helm install prom prometheus-community/kube-prometheus-stack -n monitoring --create-namespace
  kubectl get va -A -w              # watch optimization decisions
  kubectl -n ${NS} logs deploy/workload-variant-autoscaler -f

Generate load against the emulator:
  kubectl port-forward svc/vllme 8000:8000 &
  python -m inferno_amd.emulator.loadgen --url http://127.0.0.1:8000 \\
      --model default/default --schedule "[[60, 120]]"

Tear down: kind delete cluster --name ${CLUSTER}
DONE
