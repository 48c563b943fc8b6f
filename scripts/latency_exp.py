"""Per-stage latency decomposition on the GPU box: trivial vs numpy-CPU vs
numpy-GPU workloads, sequential vs concurrent, to locate the overhead."""

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

sys.path.insert(0, str(REPO / "code_interpreter_amd" / "ops"))

TRIVIAL = "print('x')"
NUMPY_TOUCH = "import numpy\nprint(numpy.__version__)"
NUMPY_GPU = (
    "import numpy\n"
    "x = numpy.random.rand(10**8)\n"
    "print(float(numpy.sum(numpy.square(x))))\n"
)
NUMPY_SMALL_GPU = (
    "import numpy\n"
    "x = numpy.random.rand(3_000_000)\n"
    "print(float(numpy.sum(numpy.square(x))))\n"
)


async def series(ex, name, code, n, conc):
    sem = asyncio.Semaphore(conc)
    lat = []

    stages = {}
    rows = []

    async def one():
        async with sem:
            t = time.perf_counter()
            r = await ex.execute(code)
            assert r.exit_code == 0, r.stderr[:300]
            wall = time.perf_counter() - t
            lat.append(wall)
            flat = {}
            for k, v in (r.timings or {}).items():
                if isinstance(v, dict):
                    flat.update(v)
                else:
                    flat[k] = v
            rows.append((wall, flat))
            for k, v in flat.items():
                stages.setdefault(k, []).append(v)

    t0 = time.perf_counter()
    await asyncio.gather(*(one() for _ in range(n)))
    dt = time.perf_counter() - t0
    lat.sort()
    print(
        f"{name:18s} conc={conc:2d}: {n / dt:6.1f} req/s  "
        f"p50={statistics.median(lat) * 1000:6.0f} ms  "
        f"min={lat[0] * 1000:5.0f} ms  max={lat[-1] * 1000:6.0f} ms  "
        + "  ".join(
            f"{k}={statistics.median(v):.1f}" for k, v in sorted(stages.items())
        ),
        flush=True,
    )
    rows.sort(key=lambda r: -r[0])
    for wall, flat in rows[:3]:
        if wall * 1000 > 3 * statistics.median(lat) * 1000 + 50:
            print(f"   WORST {wall*1000:7.0f} ms: "
                  + "  ".join(f"{k}={v}" for k, v in sorted(flat.items())),
                  flush=True)


def _print_daemon_stats(ex):
    import hipnp  # noqa (ops dir on path via package __init__)

    for i, eng in enumerate(ex._engines):
        if eng is None:
            continue
        sock = os.path.join(eng.root, "sandboxes", "gpu.sock")
        if not os.path.exists(sock):
            continue
        try:
            b = hipnp.RemoteBackend(sock)
            print(f"engine {i} daemon mem_info: {b.mem_info()}", flush=True)
        except Exception as e:
            print(f"engine {i} stats failed: {e}", flush=True)


async def main():
    tmp = tempfile.mkdtemp()
    hip = os.environ.get("EXP_HIP", "require")
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=12,
        engines_per_gpu=int(os.environ.get("EXP_ENGINES", "1")),
        gpu_count=1 if hip != "off" else 0,
        gpu_pinning=False,
        executor_root=os.path.join(tmp, "e"),
        hip_numpy=hip,
        dep_install=False,
    )
    try:
        await ex.execute(TRIVIAL)
        await asyncio.sleep(2.0)
        await series(ex, "trivial", TRIVIAL, 20, 1)
        await series(ex, "trivial", TRIVIAL, 40, 8)
        await series(ex, "numpy-touch", NUMPY_TOUCH, 20, 1)
        await series(ex, "numpy-gpu-small", NUMPY_SMALL_GPU, 20, 1)
        await series(ex, "numpy-gpu-1e8", NUMPY_GPU, 20, 1)
        await series(ex, "numpy-gpu-1e8", NUMPY_GPU, 40, 8)
        await series(ex, "numpy-gpu-small", NUMPY_SMALL_GPU, 40, 8)
        await series(ex, "sustained-1e8", NUMPY_GPU, 200, 8)
        _print_daemon_stats(ex)
        await series(ex, "sustained-trivial", TRIVIAL, 200, 8)
    finally:
        await ex.aclose()


if __name__ == "__main__":
    asyncio.run(main())
