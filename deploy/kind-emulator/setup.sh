#!/usr/bin/env bash
# Bring up a Kind cluster with EMULATED MI355X GPUs (labels + extended
# resources; no hardware needed) and the emulator-backed autoscaler stack.
# Counterpart of the reference's deploy/kind-emulator/setup.sh, AMD-first:
#   -n NODES   worker nodes (default 3)
#   -g GPUS    emulated MI355X GPUs per node (default 8)
#   -t TYPE    gpu vendor mix: amd|mix (default amd)
set -euo pipefail

NODES=3
GPUS=8
TYPE=amd
while getopts "n:g:t:" opt; do
  case $opt in
    n) NODES=$OPTARG ;;
    g) GPUS=$OPTARG ;;
    t) TYPE=$OPTARG ;;
    *) exit 1 ;;
  esac
done

CLUSTER=wva-amd
cfg=$(mktemp)
{
  echo "kind: Cluster"
  echo "apiVersion: kind.x-k8s.io/v1alpha4"
  echo "nodes:"
  echo "  - role: control-plane"
  for i in $(seq 1 "$NODES"); do echo "  - role: worker"; done
} > "$cfg"
kind create cluster --name "$CLUSTER" --config "$cfg"

# fake MI355X capacity: product labels + amd.com/gpu extended resource
i=0
for node in $(kubectl get nodes -o name | grep worker); do
  name=${node#node/}
  vendor=amd
  if [[ "$TYPE" == "mix" && $((i % 2)) == 1 ]]; then vendor=emulated; fi
  kubectl label "$node" \
    amd.com/gpu.family=CDNA4 \
    amd.com/gpu.product=MI355X \
    amd.com/gpu.vram=294912 --overwrite
  kubectl proxy --port=8001 >/dev/null 2>&1 &
  proxy_pid=$!
  sleep 1
  curl -s --header "Content-Type: application/json-patch+json" \
    --request PATCH \
    "http://127.0.0.1:8001/api/v1/nodes/$name/status" \
    --data "[{\"op\": \"add\", \"path\": \"/status/capacity/amd.com~1gpu\", \"value\": \"$GPUS\"}]" >/dev/null
  kill "$proxy_pid" 2>/dev/null || true
  i=$((i + 1))
done

echo "emulated MI355X Kind cluster '$CLUSTER' ready ($NODES nodes x $GPUS GPUs)"
echo "next: deploy/install.sh, then deploy/emulator/vllm-emulator.yaml"
