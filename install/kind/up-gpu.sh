#!/usr/bin/env bash
# GPU-enabled kind bring-up for MI355X nodes: ROCm runtime + AMD device
# plugin (replaces the reference's nvidia runtime workaround + gpu-operator,
# install/kind/up-gpu.sh). Requires /dev/kfd + /dev/dri on the host.
set -euo pipefail
"$(dirname "$0")/up.sh"
kubectl -n kube-system rollout status ds/amdgpu-device-plugin --timeout=120s
kubectl get nodes -o jsonpath='{.items[*].status.allocatable.amd\.com/gpu}'
