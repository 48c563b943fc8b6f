"""One sandboxed-torch case with step tracing (segfault isolation)."""

import asyncio
import os
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from code_interpreter_amd.services.local_executor import LocalPoolExecutor  # noqa: E402
from code_interpreter_amd.services.storage import Storage  # noqa: E402

SRC = """import torch
x = torch.randn(2048, 2048, device='cuda', dtype=torch.bfloat16)
c = x @ x
torch.cuda.synchronize()
print('ok', float(c.float().abs().sum()))
"""

# step-by-step variant: which call inside the routed path crashes the
# forked child? (route must be "off" so the import hook stays out)
SRC_STEPS = """import faulthandler, sys
faulthandler.enable()
import os
import torch
print('t1 torch imported', flush=True)
sys.path.insert(0, os.environ['APP_OPS_DIR'])
import _hipgemm
print('t2 _hipgemm imported', flush=True)
print('t3 is_available', _hipgemm.is_available(), flush=True)
x = torch.randn(1024, 1024, device='cuda', dtype=torch.bfloat16)
torch.cuda.synchronize()
print('t4 torch cuda works', flush=True)
print('t5 256ok', _hipgemm.gemm_bf16_256_ok(1024, 1024, 1024), flush=True)
c = torch.empty((1024, 1024), dtype=torch.bfloat16, device='cuda')
stream = torch.cuda.current_stream().cuda_stream
_hipgemm.gemm_raw(x.data_ptr(), x.data_ptr(), c.data_ptr(), 0, 1024, 1024, 1024, 2, stream)
torch.cuda.synchronize()
print('t6 gemm_raw done', float(c.float().abs().sum()), flush=True)
import hiptorch
print('t7 hiptorch imported', flush=True)
hiptorch.install(mode='require')
print('t8 installed', flush=True)
y = x @ x
torch.cuda.synchronize()
print('t9 routed matmul', float(y.float().abs().sum()), flush=True)
"""


async def main():
    route = sys.argv[1] if len(sys.argv) > 1 else "off"
    zygote = "--cold" not in sys.argv
    tmp = tempfile.mkdtemp(prefix="tc-")
    print("step2 tmp", tmp, "route", route, "zygote", zygote, flush=True)
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=1,
        engines_per_gpu=1,
        executor_root=os.path.join(tmp, "e"),
        dep_install=False,
        execute_timeout=240.0,
        zygote_enabled=zygote,
        hip_numpy="off" if "--nonp" in sys.argv else "auto",
    )
    print("step3 executor up", flush=True)
    src = SRC_STEPS if "--steps" in sys.argv else SRC
    r = await ex.execute(src, env={"APP_HIP_TORCH": route})
    print(
        "step4 result exit=", r.exit_code,
        "stdout=", repr(r.stdout[:120]),
        "stderr=", repr(r.stderr[-1500:]),
        flush=True,
    )
    await ex.aclose()
    print("step5 closed", flush=True)


if __name__ == "__main__":
    asyncio.run(main())
    print("step6 done", flush=True)
