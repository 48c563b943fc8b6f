"""Soak: thousands of sandboxed GPU executions; asserts steady latency and
no device-memory growth (daemon mem_info before/after)."""

import asyncio
import os
import statistics
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
sys.path.insert(0, str(REPO / "code_interpreter_amd" / "ops"))

from code_interpreter_amd.services.local_executor import LocalPoolExecutor  # noqa
from code_interpreter_amd.services.storage import Storage  # noqa

WORKLOAD = (
    "import numpy\n"
    "x = numpy.random.rand(10**8)\n"
    "print(float(numpy.sum(numpy.square(x))))\n"
)
N = int(os.environ.get("SOAK_N", "3000"))


def daemon_stats(ex):
    import hipnp

    out = []
    for eng in ex._engines:
        if eng is None:
            continue
        sock = os.path.join(eng.root, "sandboxes", "gpu.sock")
        if os.path.exists(sock):
            out.append(hipnp.RemoteBackend(sock).mem_info())
    return out


async def main():
    tmp = tempfile.mkdtemp()
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=8,
        engines_per_gpu=3,
        gpu_count=1,
        gpu_pinning=False,
        executor_root=os.path.join(tmp, "e"),
        hip_numpy="require",
        dep_install=False,
    )
    try:
        await ex.fill_pool()
        await ex.execute("print('warm')")
        await asyncio.sleep(2.0)
        print("daemon mem before:", daemon_stats(ex), flush=True)

        lat = []
        errors = 0
        sem = asyncio.Semaphore(12)

        async def one(i):
            nonlocal errors
            async with sem:
                t = time.perf_counter()
                r = await ex.execute(WORKLOAD)
                if r.exit_code != 0:
                    errors += 1
                lat.append(time.perf_counter() - t)

        t0 = time.perf_counter()
        await asyncio.gather(*(one(i) for i in range(N)))
        dt = time.perf_counter() - t0
        lat.sort()
        half = len(lat) // 2
        print(
            f"soak {N} requests: {N / dt:.0f} req/s  errors={errors}  "
            f"p50={statistics.median(lat) * 1e3:.1f} ms  "
            f"p99={lat[int(len(lat) * 0.99) - 1] * 1e3:.1f} ms  "
            f"first-half p50={statistics.median(lat[:half]) * 1e3:.1f}",
            flush=True,
        )
        print("daemon mem after:", daemon_stats(ex), flush=True)
        assert errors == 0
    finally:
        await ex.aclose()


if __name__ == "__main__":
    asyncio.run(main())
