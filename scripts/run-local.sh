#!/usr/bin/env bash
# Build native components and run the service locally (no k8s): HTTP on
# :50081, gRPC on :50051, sandboxes fanned across all visible GPUs.
set -euo pipefail
cd "$(dirname "$0")/.."
make -C code_interpreter_amd/executor
python -m code_interpreter_amd.ops.build
exec python -m code_interpreter_amd
