#!/usr/bin/env bash
# kind with AMD GPUs: mount /dev/kfd + /dev/dri into the node and install
# the ROCm k8s-device-plugin so pods can request amd.com/gpu
# (replaces the reference's nvidia gpu-operator flow, install/kind/up-gpu.sh).
set -euo pipefail
kind create cluster --name substratus-gpu --config - <<'KINDCFG'
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
  - hostPath: /dev/kfd
    containerPath: /dev/kfd
  - hostPath: /dev/dri
    containerPath: /dev/dri
KINDCFG
# ROCm device plugin DaemonSet (publishes amd.com/gpu)
kubectl create -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-dp.yaml
kubectl apply -f ../../config/crd/bases
kubectl apply -k ../../config/registry-kind
kubectl apply -k ../../config/install-kind
