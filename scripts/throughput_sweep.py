"""Sustained-throughput sweep over engines x concurrency (GPU box)."""

import asyncio
import os
import statistics
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from code_interpreter_amd.services.local_executor import LocalPoolExecutor  # noqa
from code_interpreter_amd.services.storage import Storage  # noqa

WORKLOAD = (
    "import numpy\n"
    "x = numpy.random.rand(10**8)\n"
    "print(float(numpy.sum(numpy.square(x))))\n"
)


async def measure(engines: int, conc: int, requests: int = 300):
    tmp = tempfile.mkdtemp()
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=max(4, conc // engines),
        engines_per_gpu=engines,
        gpu_count=1,
        gpu_pinning=False,
        executor_root=os.path.join(tmp, "e"),
        hip_numpy="require",
        dep_install=False,
    )
    try:
        await ex.fill_pool()
        await ex.execute("print('warm')")
        await asyncio.sleep(2.5)
        lat = []
        sem = asyncio.Semaphore(conc)

        async def one():
            async with sem:
                t = time.perf_counter()
                r = await ex.execute(WORKLOAD)
                assert r.exit_code == 0, r.stderr[:300]
                lat.append(time.perf_counter() - t)

        t0 = time.perf_counter()
        await asyncio.gather(*(one() for _ in range(requests)))
        dt = time.perf_counter() - t0
        lat.sort()
        print(
            f"engines={engines} conc={conc:3d}: {requests / dt:6.1f} req/s  "
            f"p50={statistics.median(lat) * 1000:5.0f} ms  "
            f"p95={lat[int(len(lat) * 0.95) - 1] * 1000:5.0f} ms",
            flush=True,
        )
    finally:
        await ex.aclose()


async def main():
    for engines, conc in ((2, 8), (2, 16), (3, 12), (4, 16), (4, 24), (6, 24)):
        await measure(engines, conc)


if __name__ == "__main__":
    asyncio.run(main())
