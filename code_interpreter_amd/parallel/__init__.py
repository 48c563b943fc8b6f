"""Multi-GPU parallel execution over RCCL/xGMI.

Re-exports the sandbox-side helpers (ops/mgpu.py): one process per GPU
with torch.distributed backend "nccl" (RCCL on ROCm), xGMI-sized bucketed
all-reduce, and the BASELINE matmul+all-reduce acceptance workload.
"""

import os
import sys

_OPS_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "ops")
if _OPS_DIR not in sys.path:
    sys.path.append(_OPS_DIR)

from mgpu import (  # noqa: E402,F401
    DEFAULT_BUCKET_BYTES,
    allreduce_bucketed,
    allreduce_matmul_bench,
    run_distributed,
    visible_gpu_count,
)

from code_interpreter_amd.parallel.topology import (  # noqa: E402,F401
    XGMI_LINK_GBPS,
    XGMI_LINKS_PER_GPU,
    gpu_inventory,
    recommended_bucket_bytes,
    ring_allreduce_seconds,
)

__all__ = [
    "XGMI_LINK_GBPS",
    "XGMI_LINKS_PER_GPU",
    "gpu_inventory",
    "recommended_bucket_bytes",
    "ring_allreduce_seconds",
    "DEFAULT_BUCKET_BYTES",
    "allreduce_bucketed",
    "allreduce_matmul_bench",
    "run_distributed",
    "visible_gpu_count",
]
