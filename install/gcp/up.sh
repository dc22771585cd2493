#!/usr/bin/env bash
# GKE install (parity: reference install/gcp/up.sh) targeting MI355X node
# pools. Requires: gcloud auth, PROJECT set.
set -euo pipefail
export PROJECT=${PROJECT:-$(gcloud config get-value project)}
export CLUSTER_NAME=${CLUSTER_NAME:-substratus}
export REGION=${REGION:-us-central1}
gcloud container clusters create "$CLUSTER_NAME" --region "$REGION" \
  --enable-autoprovisioning --max-cpu 512 --max-memory 4096 \
  --addons GcsFuseCsiDriver --workload-pool "$PROJECT.svc.id.goog"
# MI355X node pool (amd.com/gpu via the ROCm device plugin DaemonSet)
gcloud container node-pools create mi355x --cluster "$CLUSTER_NAME" \
  --region "$REGION" --machine-type a4x-highgpu-8g-amd --num-nodes 1 || true
kubectl create -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-dp.yaml
kubectl create -f https://raw.githubusercontent.com/ROCm/k8s-device-plugin/master/k8s-ds-amdgpu-labeller.yaml
gsutil mb "gs://$PROJECT-substratus-artifacts" || true
gcloud artifacts repositories create substratus --repository-format=docker \
  --location "$REGION" || true
gcloud iam service-accounts create substratus || true
kubectl apply -f ../../config/crd/bases
kubectl apply -k ../../config/install-gcp
