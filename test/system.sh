#!/usr/bin/env bash
# E2E smoke on kind (parity: reference test/system.sh): bring up the
# control plane, apply the opt-125m Model + Server, wait ready, and curl
# /v1/completions through a port-forward.
set -euo pipefail
cd "$(dirname "$0")/.."
bash install/kind/up.sh
kubectl wait --for=condition=Available deployment/controller-manager \
  -n substratus --timeout=300s
kubectl apply -f examples/facebook-opt-125m/base-model.yaml
kubectl apply -f examples/facebook-opt-125m/base-server.yaml
kubectl wait --for=jsonpath='{.status.ready}'=true model/facebook-opt-125m \
  --timeout=600s
kubectl wait --for=jsonpath='{.status.ready}'=true server/facebook-opt-125m \
  --timeout=600s
kubectl port-forward service/facebook-opt-125m-server 8080:8080 &
PF=$!
trap "kill $PF" EXIT
sleep 3
curl -sf http://localhost:8080/v1/completions \
  -H 'Content-Type: application/json' \
  -d '{"prompt": "Kubernetes is", "max_tokens": 8}' | tee /dev/stderr \
  | grep -q text_completion
echo "SYSTEM TEST PASSED"
