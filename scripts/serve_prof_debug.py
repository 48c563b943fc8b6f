"""Debug helper: one engine + one GPU-routed execution, then a graceful
close — used under rocprofv3 to verify the daemon inherits the profiler
environment and flushes kernel output on SIGTERM."""

import asyncio
import os
import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from code_interpreter_amd.services.local_executor import LocalPoolExecutor
from code_interpreter_amd.services.storage import Storage


async def main():
    tmp = tempfile.mkdtemp(prefix="prof-dbg-")
    ex = LocalPoolExecutor(
        Storage(os.path.join(tmp, "s")),
        pool_target_length=1,
        engines_per_gpu=1,
        executor_root=os.path.join(tmp, "e"),
        dep_install=False,
    )
    code = (
        "import numpy, hipnp\n"
        "x = numpy.random.rand(20_000_000)\n"
        "print(hipnp.backend().name, float(numpy.sum(numpy.square(x))))\n"
    )
    for i in range(5):
        r = await ex.execute(code)
        print("exec", i, r.exit_code, r.stdout.strip()[:60])

    # dump the daemon's profiler-relevant environment
    import psutil

    eng = ex._engines[0]
    for child in psutil.Process(eng.proc.pid).children(recursive=True):
        cmd = " ".join(child.cmdline())
        if "hipd.py" in cmd:
            env = child.environ()
            keys = [k for k in env if "ROCP" in k or "PRELOAD" in k or "HSA_TOOLS" in k]
            print("daemon pid", child.pid, "profiler env:")
            for k in keys:
                print("   ", k, "=", env[k][:120])
    await ex.aclose()
    print("closed gracefully")


asyncio.run(main())
