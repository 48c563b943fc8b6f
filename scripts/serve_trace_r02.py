"""r02 serving-path trace workload: sandboxed executions exercising the
round-2 device surface (axis reductions, boolean masks, np.where,
broadcasting, argmax) through the real engine + GPU daemon, for a
rocprofv3 kernel-stats capture of the daemon process.

    rocprofv3 --kernel-trace --stats --output-format csv -d OUT -- \
        python scripts/serve_trace_r02.py
"""

import asyncio
import os
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from code_interpreter_amd.services.local_executor import LocalPoolExecutor  # noqa
from code_interpreter_amd.services.storage import Storage  # noqa

SRC = """
import numpy
x = numpy.random.rand(4000, 5000)          # device Philox
col_mu = x.mean(axis=0)                     # axis reduce (inner kernel)
row_sum = x.sum(axis=1)                     # axis reduce (wave/slice)
centered = x - col_mu                       # broadcast binary
mask = centered > 0.25                      # compare -> u8 mask
n_hi = int(mask.sum())                      # popcount
clipped = numpy.where(mask, centered, 0.0)  # select
peak = int(numpy.argmax(row_sum))           # argminmax
total = float(numpy.sum(numpy.square(clipped)))  # fused square+sum
print("ok", n_hi, peak, round(total, 3))
"""


async def main():
    tmp = tempfile.mkdtemp(prefix="trace-r02-")
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=1,
        engines_per_gpu=1,
        executor_root=os.path.join(tmp, "e"),
        dep_install=False,
        hip_numpy="require",
    )
    try:
        for i in range(3):
            r = await ex.execute(SRC)
            assert r.exit_code == 0, r.stderr
            print("run", i, r.stdout.strip())
    finally:
        await ex.aclose()


if __name__ == "__main__":
    asyncio.run(main())
