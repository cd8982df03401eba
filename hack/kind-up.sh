#!/usr/bin/env bash
# Local kind cluster + gateway image (reference hack/ analog).
set -euo pipefail
CLUSTER=${CLUSTER:-inference-gateway}
IMAGE=${IMAGE:-inference-gateway-amd:dev}

kind get clusters | grep -qx "$CLUSTER" || kind create cluster --name "$CLUSTER"
docker build -t "$IMAGE" --target gateway .
kind load docker-image "$IMAGE" --name "$CLUSTER"
echo "cluster '$CLUSTER' ready; image '$IMAGE' loaded"
echo "next: kubectl apply -f examples/kubernetes/basic.yaml"
