#!/usr/bin/env bash
# Build native components and run the service locally (no k8s): HTTP on
# :50081, gRPC on :50051, sandboxes fanned across all visible GPUs.
set -euo pipefail
cd "$(dirname "$0")/.."
make -C code_interpreter_amd/executor
python -m code_interpreter_amd.ops.build
# measured serving optimum on a 16-CPU / 1-GPU box (profiles/NOTES.md):
# several sandbox engines per GPU keep the fork path parallel
export APP_ENGINES_PER_GPU="${APP_ENGINES_PER_GPU:-6}"
exec python -m code_interpreter_amd
