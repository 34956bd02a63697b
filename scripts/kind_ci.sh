#!/usr/bin/env bash
# Real-Kubernetes CI tier (BASELINE config 1): install the chart on a kind
# cluster and run hello_world -> reload -> teardown through K8sDriver, the
# headless-DNS discovery path and the NDJSON pod channel.
#
# Requires: kind, kubectl, helm, docker (none are available in the build
# container — run this on a workstation or CI runner with Docker).
# The fake-kubectl tier (tests/test_k8s_driver.py) covers K8sDriver
# behavior without a cluster and runs everywhere.
set -euo pipefail

CLUSTER=${CLUSTER:-kt-amd-ci}
NS=${NS:-kt-ci}
IMG=${IMG:-kubetorch-amd-worker:ci}

for bin in kind kubectl helm docker; do
  command -v "$bin" >/dev/null || { echo "SKIP: $bin not installed"; exit 0; }
done

echo "==> kind cluster"
kind get clusters | grep -q "^${CLUSTER}$" || kind create cluster --name "$CLUSTER" --wait 120s

echo "==> worker image (CPU-only torch is fine for the control-plane tier)"
docker build -t "$IMG" -f Dockerfile .
kind load docker-image "$IMG" --name "$CLUSTER"

echo "==> chart"
kubectl create namespace "$NS" --dry-run=client -o yaml | kubectl apply -f -
helm upgrade --install kubetorch-amd charts/kubetorch-amd \
  -n "$NS" --set image="$IMG" --wait --timeout 300s

echo "==> controller reachable"
kubectl -n "$NS" port-forward svc/kubetorch-controller 8080:8080 &
PF=$!
trap 'kill $PF 2>/dev/null || true' EXIT
sleep 3
curl -fsS http://127.0.0.1:8080/health

echo "==> hello_world -> reload -> teardown through the API"
KT_API_URL=http://127.0.0.1:8080 KT_NAMESPACE="$NS" KT_USERNAME=kindci \
  python - <<'PY'
import kubetorch_amd as kt

def hello(x):
    return f"hello {x}"

f = kt.fn(hello).to(kt.Compute(cpus=1))
assert f("kind") == "hello kind"
f.to()  # hot reload into the warm pod
assert f("again") == "hello again"
f.teardown()
print("kind tier OK")
PY
echo "==> PASS"
