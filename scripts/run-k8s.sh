#!/usr/bin/env bash
# Build both images, deploy to the current kubectl context, wait for
# readiness and port-forward HTTP (50081) + gRPC (50051)
# (parity: reference scripts/run-build.sh).
set -euo pipefail
cd "$(dirname "$0")/.."
docker build -f docker/Dockerfile.service -t localhost/code-interpreter-amd:local .
docker build -f docker/Dockerfile.executor -t localhost/code-interpreter-amd-executor:local .
kubectl apply -f k8s/local.yaml
kubectl wait --for=condition=Ready pod/code-interpreter-amd --timeout=300s
kubectl port-forward pod/code-interpreter-amd 50081:50081 50051:50051 &
trap 'kill %1' EXIT
kubectl logs -f pod/code-interpreter-amd
