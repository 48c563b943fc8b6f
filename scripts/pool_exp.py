"""Warm-pool scaling experiment (run on GPU box): sustained exec
throughput vs zygote warm-children pool size, plus raw child prewarm
latency. Drives LocalPoolExecutor directly (no HTTP) to isolate the
sandbox pipeline."""

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


async def measure(pool: int, concurrency: int, requests: int) -> None:
    tmp = tempfile.mkdtemp()
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=pool,
        gpu_count=1,
        gpu_pinning=False,
        executor_root=os.path.join(tmp, "e"),
        hip_numpy="require",
        dep_install=False,
    )
    try:
        t0 = time.perf_counter()
        await ex.execute("print('warm')")
        first = time.perf_counter() - t0

        # let the pool warm up fully
        await asyncio.sleep(3.0)

        lat = []
        sem = asyncio.Semaphore(concurrency)

        async def one():
            async with sem:
                t = time.perf_counter()
                r = await ex.execute(WORKLOAD)
                assert r.exit_code == 0, r.stderr[:400]
                lat.append(time.perf_counter() - t)

        t0 = time.perf_counter()
        await asyncio.gather(*(one() for _ in range(requests)))
        dt = time.perf_counter() - t0
        lat.sort()
        print(
            f"pool={pool:3d} conc={concurrency:3d}: {requests / dt:6.1f} req/s  "
            f"p50={statistics.median(lat) * 1000:6.0f} ms  "
            f"p95={lat[int(len(lat) * 0.95) - 1] * 1000:6.0f} ms  "
            f"(first exec {first * 1000:.0f} ms)",
            flush=True,
        )
    finally:
        await ex.aclose()


async def main():
    # raw prewarm latency: engine spawn -> first warm child ready
    for pool, conc in ((6, 6), (12, 6), (16, 12), (32, 16), (48, 32)):
        await measure(pool, conc, max(32, conc * 4))


if __name__ == "__main__":
    asyncio.run(main())
