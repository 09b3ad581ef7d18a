#!/usr/bin/env bash
# Install the MI355X-native workload-variant autoscaler.
#
# Two modes (the counterpart of the reference's deploy/install.sh — Kind
# bring-up with emulated GPUs, Prometheus stack, WVA and sample workload):
#
#   --local     Bring up the full LOCAL stack with no cluster binaries:
#               kube-apiserver stand-in + vLLM emulator + TLS Prometheus
#               stand-in + the real controller process, with the CRD, the
#               three ConfigMaps, a variant Deployment and the sample VA
#               applied. Works in any box with this repo's Python env.
#               Add --smoke to drive load and assert scale-out, then exit.
#
#   (default)   Install into the current kubectl context: CRD, namespace,
#               ConfigMaps, RBAC + controller Deployment, the vLLM emulator
#               Deployment/Service, and optionally the sample VA + HPA.
#               Requires kubectl; Prometheus is expected at
#               $PROMETHEUS_BASE_URL (kube-prometheus-stack or equivalent).
#
# Flags (cluster mode):
#   --namespace NS      controller namespace (default workload-variant-autoscaler-system)
#   --with-emulator     deploy the vLLM emulator + Service (default on)
#   --no-emulator       skip the emulator
#   --with-sample-va    apply deploy/examples/vllme-variantautoscaling.yaml
#   --with-hpa          apply deploy/examples/hpa-integration.yaml
#   --image IMG         controller image (default inferno-amd/controller:latest)
#   --uninstall         delete everything this script created
set -euo pipefail

SCRIPT_DIR="$(cd "$(dirname "${BASH_SOURCE[0]}")" && pwd)"
REPO_DIR="$(dirname "$SCRIPT_DIR")"
NAMESPACE="workload-variant-autoscaler-system"
MODE="cluster"
SMOKE=""
WITH_EMULATOR=1
WITH_SAMPLE_VA=0
WITH_HPA=0
UNINSTALL=0
IMAGE="inferno-amd/controller:latest"

while [[ $# -gt 0 ]]; do
  case "$1" in
    --local) MODE="local" ;;
    --smoke) SMOKE="--smoke" ;;
    --namespace) NAMESPACE="$2"; shift ;;
    --with-emulator) WITH_EMULATOR=1 ;;
    --no-emulator) WITH_EMULATOR=0 ;;
    --with-sample-va) WITH_SAMPLE_VA=1 ;;
    --with-hpa) WITH_HPA=1 ;;
    --image) IMAGE="$2"; shift ;;
    --uninstall) UNINSTALL=1 ;;
    -h|--help) grep '^#' "$0" | sed 's/^# \{0,1\}//'; exit 0 ;;
    *) echo "unknown flag: $1" >&2; exit 2 ;;
  esac
  shift
done

if [[ "$MODE" == "local" ]]; then
  cd "$REPO_DIR"
  if [[ -n "$SMOKE" ]]; then
    exec python3 -m inferno_amd.testing.stack --smoke --interval 2s
  fi
  exec python3 -m inferno_amd.testing.stack
fi

command -v kubectl >/dev/null || {
  echo "kubectl not found — for a cluster-free bring-up use: $0 --local" >&2
  exit 1
}

if [[ "$UNINSTALL" == 1 ]]; then
  kubectl delete -f "$SCRIPT_DIR/examples/hpa-integration.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/examples/vllme-variantautoscaling.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/emulator.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/controller.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/configmap-serviceclass.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/configmap-accelerator-unitcost.yaml" --ignore-not-found || true
  kubectl delete -f "$SCRIPT_DIR/crd/llmd.ai_variantautoscalings.yaml" --ignore-not-found || true
  kubectl delete namespace "$NAMESPACE" --ignore-not-found || true
  echo "uninstalled."
  exit 0
fi

echo ">>> CRD"
kubectl apply -f "$SCRIPT_DIR/crd/llmd.ai_variantautoscalings.yaml"

echo ">>> namespace $NAMESPACE"
kubectl create namespace "$NAMESPACE" --dry-run=client -o yaml | kubectl apply -f -

echo ">>> ConfigMaps (accelerator costs, service classes, WVA config)"
kubectl apply -f "$SCRIPT_DIR/configmap-accelerator-unitcost.yaml"
kubectl apply -f "$SCRIPT_DIR/configmap-serviceclass.yaml"
kubectl -n "$NAMESPACE" create configmap \
  workload-variant-autoscaler-variantautoscaling-config \
  --from-literal=GLOBAL_OPT_INTERVAL="${GLOBAL_OPT_INTERVAL:-60s}" \
  --from-literal=PROMETHEUS_BASE_URL="${PROMETHEUS_BASE_URL:-https://prometheus-k8s.monitoring.svc.cluster.local:9090}" \
  --dry-run=client -o yaml | kubectl apply -f -

echo ">>> controller (RBAC + Deployment, image $IMAGE)"
sed "s|inferno-amd/controller:latest|$IMAGE|" "$SCRIPT_DIR/controller.yaml" | kubectl apply -f -

if [[ "$WITH_EMULATOR" == 1 ]]; then
  echo ">>> vLLM emulator"
  kubectl apply -f "$SCRIPT_DIR/emulator.yaml"
fi

if [[ "$WITH_SAMPLE_VA" == 1 ]]; then
  echo ">>> sample VariantAutoscaling"
  kubectl apply -f "$SCRIPT_DIR/examples/vllme-variantautoscaling.yaml"
fi

if [[ "$WITH_HPA" == 1 ]]; then
  echo ">>> HPA integration sample"
  kubectl apply -f "$SCRIPT_DIR/examples/hpa-integration.yaml"
fi

echo ">>> waiting for controller rollout"
kubectl -n "$NAMESPACE" rollout status deployment/workload-variant-autoscaler \
  --timeout=180s || true

echo "done. Verify with:"
echo "  kubectl get variantautoscalings -A"
echo "  kubectl -n $NAMESPACE logs deploy/workload-variant-autoscaler"
