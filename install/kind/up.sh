#!/usr/bin/env bash
# kind cluster bring-up (reference install/kind/up.sh): extraPortMapping
# 30080 exposes the SCI signed-URL NodePort to the host.
set -euo pipefail
kind create cluster --name substratus --config - <<'KIND'
kind: Cluster
apiVersion: kind.x-k8s.io/v1alpha4
nodes:
- role: control-plane
  extraPortMappings:
  - containerPort: 30080
    hostPort: 30080
  extraMounts:
  - hostPath: /tmp/substratus-bucket
    containerPath: /bucket
KIND
kubectl apply -f "$(dirname "$0")/../../config/crd/bases/"
kubectl apply -f "$(dirname "$0")/../kubernetes/kind/system.yaml"
