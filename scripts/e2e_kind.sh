#!/usr/bin/env bash
# Kind-cluster e2e for BASELINE config #1: operator + gateway + a CPU-runtime
# ArksApplication on a local kind cluster (the reference's test/e2e analogue).
# Requires: kind, kubectl, docker. The in-container CI analogue (no docker)
# is tests/test_e2e_local.py — this script is the real-cluster leg.
set -euo pipefail

CLUSTER=${CLUSTER:-arks-e2e}
IMG=${IMG:-arks-amd/runtime:e2e}

echo ">> building runtime image"
docker build -t "$IMG" .

echo ">> creating kind cluster $CLUSTER"
kind get clusters | grep -qx "$CLUSTER" || kind create cluster --name "$CLUSTER"
kind load docker-image "$IMG" --name "$CLUSTER"

echo ">> installing CRDs + operator + gateway"
kubectl apply -f deploy/crds/
kubectl create namespace arks-system --dry-run=client -o yaml | kubectl apply -f -
sed "s|arks-amd/runtime:latest|$IMG|" deploy/operator.yaml | kubectl apply -f -
sed "s|arks-amd/runtime:latest|$IMG|" deploy/gateway.yaml | kubectl apply -f -
kubectl -n arks-system rollout status deploy/arks-operator --timeout=180s
kubectl -n arks-system rollout status deploy/arks-gateway --timeout=180s

echo ">> applying the quickstart sample (CPU tiny runtime override)"
kubectl apply -f deploy/samples/quickstart.yaml

echo ">> waiting for the application to reconcile"
for i in $(seq 1 60); do
  PHASE=$(kubectl get arksapplication qwen-app -o jsonpath='{.status.phase}' 2>/dev/null || true)
  echo "  phase=$PHASE"
  [ "$PHASE" = "Running" ] && break
  sleep 5
done

echo ">> curling a completion through the gateway"
kubectl -n arks-system port-forward svc/arks-gateway 18080:80 &
PF=$!
trap 'kill $PF 2>/dev/null || true' EXIT
sleep 2
curl -sf http://127.0.0.1:18080/v1/chat/completions \
  -H "Authorization: Bearer team-a-secret-token" \
  -H "Content-Type: application/json" \
  -d '{"model":"qwen2.5-7b-instruct","messages":[{"role":"user","content":"hi"}],"max_tokens":4}' \
  | tee /dev/stderr | grep -q '"choices"'
echo ">> e2e OK"
