#!/usr/bin/env bash
# Install the MI355X workload-variant autoscaler into a cluster.
# Prereqs: a reachable HTTPS Prometheus (the controller refuses to start
# without one) and, for actuation, HPA + prometheus-adapter or KEDA.
set -euo pipefail

NS=workload-variant-autoscaler-system
DIR="$(cd "$(dirname "${BASH_SOURCE[0]}")" && pwd)"

kubectl apply -f "$DIR/crd/llmd.ai_variantautoscalings.yaml"
kubectl create namespace "$NS" --dry-run=client -o yaml | kubectl apply -f -
kubectl apply -f "$DIR/configmap-controller.yaml"
kubectl apply -f "$DIR/configmap-accelerator-unitcost.yaml"
kubectl apply -f "$DIR/configmap-service-classes.yaml"
kubectl apply -f "$DIR/controller.yaml"

echo "waiting for controller rollout..."
kubectl -n "$NS" rollout status deployment/wva-amd-controller --timeout=180s

cat <<'MSG'
Installed. Next steps:
  - deploy a model server (or the emulator: deploy/emulator/vllm-emulator.yaml)
  - create a VariantAutoscaling (deploy/samples/mi355x-variantautoscaling.yaml)
  - wire actuation: deploy/integrations/hpa.yaml or keda-scaledobject.yaml
MSG
