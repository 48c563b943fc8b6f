"""Failure-recovery tests: engine crashes are absorbed by respawn +
whole-execution retry (the reference's equivalent surface is pod-spawn
retry + k8s GC, kubernetes_code_executor.py:75-79, 191-195)."""

import asyncio
import os
import signal

import pytest

from code_interpreter_amd.services.local_executor import LocalPoolExecutor
from code_interpreter_amd.services.storage import Storage


@pytest.fixture
def executor(tmp_path, executor_bin):
    ex = LocalPoolExecutor(
        Storage(str(tmp_path / "s")),
        pool_target_length=1,
        gpu_count=0,
        executor_root=str(tmp_path / "e"),
        dep_install=False,
    )
    yield ex
    asyncio.run(ex.aclose())


def test_engine_crash_recovers(executor):
    async def run():
        r = await executor.execute("print('before')")
        assert r.exit_code == 0
        # kill the engine out from under the executor
        engine = executor._engines[0]
        os.killpg(engine.proc.pid, signal.SIGKILL)
        engine.proc.wait()
        # next execution must respawn and succeed (retry layer)
        r = await executor.execute("print('after')")
        assert r.exit_code == 0
        assert r.stdout == "after\n"

    asyncio.run(run())


def test_user_timeout_contract(executor):
    """Hanging user code is bounded by the executor timeout and reported
    with the reference's exact shape ('', 'Execution timed out', -1)."""
    executor.execute_timeout = 2.0

    async def run():
        r = await executor.execute("import time\ntime.sleep(60)")
        assert r.exit_code == -1
        assert r.stderr == "Execution timed out"
        assert r.stdout == ""
        # the engine survives a timed-out execution
        r = await executor.execute("print('alive')")
        assert r.stdout == "alive\n"

    asyncio.run(run())


def test_crash_in_user_code_is_contained(executor):
    async def run():
        r = await executor.execute(
            "import ctypes\nctypes.string_at(0)"  # segfault the sandbox
        )
        assert r.exit_code != 0
        r = await executor.execute("print('still up')")
        assert r.stdout == "still up\n"

    asyncio.run(run())


def test_zygote_crash_recovers(executor):
    """Killing the engine's zygote degrades to cold fork/exec and the
    monitor respawns it; executions keep succeeding throughout."""
    import time

    import psutil

    async def run():
        r = await executor.execute("print('warm path')")
        assert r.exit_code == 0
        engine = executor._engines[0]
        server = psutil.Process(engine.proc.pid)
        def zygote_children():
            out = []
            for c in server.children():
                try:
                    if "zygote.py" in " ".join(c.cmdline()):
                        out.append(c)
                except (psutil.ZombieProcess, psutil.NoSuchProcess):
                    continue
            return out

        zygotes = zygote_children()
        assert zygotes, "zygote process not found"
        for z in zygotes:
            z.kill()
        # immediately after the kill: cold path must serve
        r = await executor.execute("print('cold path')")
        assert r.exit_code == 0 and r.stdout == "cold path\n"
        # give the monitor time to respawn, then execute again
        time.sleep(2.5)
        r = await executor.execute("print('respawned')")
        assert r.exit_code == 0 and r.stdout == "respawned\n"
        assert zygote_children(), "zygote was not respawned"

    asyncio.run(run())

def test_infra_exit_code_triggers_whole_execution_retry(executor, tmp_path):
    """Exit code 113 + the GpuBackendLost stderr marker (emitted by
    sandbox_runtime when the GPU daemon dies mid-script with live device
    handles) must be treated as an infrastructure failure: the executor
    retries the whole execution in a fresh sandbox instead of returning
    the failure to the user."""
    marker = tmp_path / "first_attempt_done"
    code = (
        "import os, sys\n"
        f"m = {str(marker)!r}\n"
        "if not os.path.exists(m):\n"
        "    open(m, 'w').close()\n"
        "    print('gpu backend lost; execution will be retried',"
        " file=sys.stderr)\n"
        "    sys.exit(113)\n"
        "print('second attempt ok')\n"
    )

    async def run():
        r = await executor.execute(code)
        assert r.exit_code == 0, (r.exit_code, r.stderr)
        assert r.stdout == "second attempt ok\n"

    asyncio.run(run())


def test_backpressure_queues_overload(tmp_path, executor_bin):
    """With a 1-deep in-flight cap, a burst of concurrent executions is
    serialized through the queue -- all succeed, none rejected."""
    ex = LocalPoolExecutor(
        Storage(str(tmp_path / "s2")),
        pool_target_length=1,
        gpu_count=0,
        executor_root=str(tmp_path / "e2"),
        dep_install=False,
        max_inflight_per_engine=1,
    )

    async def run():
        results = await asyncio.gather(
            *(ex.execute(f"print({i})") for i in range(12))
        )
        assert [r.exit_code for r in results] == [0] * 12
        assert sorted(r.stdout for r in results) == sorted(
            f"{i}\n" for i in range(12)
        )

    try:
        asyncio.run(run())
    finally:
        asyncio.run(ex.aclose())


def test_multi_gpu_engine_pinning_round_robin(tmp_path, executor_bin):
    """gpu_count=8 on a CPU box: engines still spawn (CPU sandboxes) and
    each engine process carries its round-robin HIP_VISIBLE_DEVICES pin —
    the data-parallel fan-out config the driver's 8-GPU run relies on."""
    import psutil

    from code_interpreter_amd.services.local_executor import LocalPoolExecutor
    from code_interpreter_amd.services.storage import Storage

    ex = LocalPoolExecutor(
        Storage(str(tmp_path / "s")),
        pool_target_length=1,
        engines_per_gpu=1,
        gpu_count=8,
        gpu_pinning=True,
        executor_root=str(tmp_path / "e"),
        server_bin=executor_bin,
        dep_install=False,
        hip_numpy="off",
    )

    async def run():
        assert ex.n_engines == 8
        # touch a few engines (full 8 would be slow on CI): indexes 0,3,7
        for idx in (0, 3, 7):
            engine = await ex._ensure_engine(idx)
            proc = psutil.Process(engine.proc.pid)
            env = proc.environ()
            assert env.get("HIP_VISIBLE_DEVICES") == str(idx % 8), (
                idx, env.get("HIP_VISIBLE_DEVICES"))
        r = await ex.execute("print('pinned-ok')")
        assert r.exit_code == 0 and "pinned-ok" in r.stdout
        await ex.aclose()

    asyncio.run(run())
