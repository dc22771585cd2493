#!/usr/bin/env bash
# Bring up a local kind cluster for the control plane (CPU plumbing path,
# parity: reference install/kind/up.sh). NodePort 30080 exposes the kind
# SCI's signed-URL HTTP handler.
set -euo pipefail
kind create cluster --name substratus --config - <<'KINDCFG'
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
KINDCFG
kubectl apply -f ../../config/crd/bases
kubectl apply -k ../../config/registry-kind
kubectl apply -k ../../config/install-kind
