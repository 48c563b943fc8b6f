"""Custom-tool torch path for rocprofv3 tracing (VERDICT r01 #1 evidence):
runs a /v1/execute-custom-tool-style tool whose body is the acceptance
torch bf16 matmul through the REAL sandbox stack (LocalPoolExecutor ->
executor-server -> zygote child -> torch + hiptorch routing), so a
kernel trace of this process tree shows which GEMM kernel the custom-tool
path actually runs (r01: hipBLASLt's Cijk_*; r02 target: our
gemm_bf16_256t_kernel).

Usage (GPU box):
    rocprofv3 --kernel-trace --stats --output-format csv -d OUT -- \
        python scripts/custom_tool_trace.py [size] [iters]
"""

import asyncio
import os
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

TOOL = '''
import torch

def bf16_matmul_bench(size: int = 8192, iters: int = 4) -> dict:
    """Time a bf16 matmul on the GPU.

    :param size: matrix dimension
    :param iters: timed iterations
    :return: timing stats
    """
    import time
    a = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    c = a @ b
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        c = a @ b
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return {"ms": dt * 1e3, "tflops": 2 * size**3 / dt / 1e12,
            "checksum": float(c.float().abs().sum())}
'''


async def main():
    size = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 4

    from code_interpreter_amd.services.custom_tool_executor import (
        CustomToolExecutor,
    )
    from code_interpreter_amd.services.local_executor import LocalPoolExecutor
    from code_interpreter_amd.services.storage import Storage

    os.environ.setdefault("APP_HIP_TORCH", "require")
    tmp = tempfile.mkdtemp(prefix="ctool-trace-")
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "storage")),
        pool_target_length=1,
        engines_per_gpu=1,
        executor_root=os.path.join(tmp, "executors"),
        dep_install=False,
        execute_timeout=300.0,
    )
    tool_ex = CustomToolExecutor(ex)
    try:
        out = await tool_ex.execute(
            TOOL, f'{{"size": {size}, "iters": {iters}}}', {}
        )
        print("tool output:", out)
    finally:
        await ex.aclose()


if __name__ == "__main__":
    asyncio.run(main())
