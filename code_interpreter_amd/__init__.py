"""MI355X-native sandboxed code-execution service.

A from-scratch rebuild of the capabilities of i-am-bee/bee-code-interpreter
(reference: /root/reference) designed for AMD Instinct MI355X (gfx950) nodes:

- asyncio control plane (HTTP + gRPC) with the reference's wire contracts
  (POST /v1/execute, /v1/parse-custom-tool, /v1/execute-custom-tool and the
  matching CodeInterpreterService gRPC API),
- warm pools of single-use sandbox executors, each pinned to one of the
  node's 8 MI355X GPUs (data-parallel request fan-out),
- a C++ in-pod executor server (replacing the reference's Rust
  executor/server.rs) with a pre-forked Python "zygote" so per-request
  interpreter start-up cost is a fork, not a cold import,
- hand-written CDNA4 (gfx950) HIP kernels (MFMA GEMM, elementwise,
  reduction, RNG) that user numpy compute is routed to, with pinned
  hipHostMalloc + hipMemcpyAsync staging,
- RCCL-over-xGMI collectives for multi-GPU user jobs.
"""

__version__ = "0.1.0"
