"""Local-process executor backend: warm pools of single-use sandboxes fanned
across the node's MI355X GPUs.

The MI355X-native equivalent of the reference's warm Kubernetes pod queue
(kubernetes_code_executor.py:151-264), with the pod replaced by a local
executor-server process (executor/server.cpp + zygote) listening on a unix
socket:

- each sandbox is pinned to one GPU via HIP_VISIBLE_DEVICES (round-robin
  across `gpu_count` devices), so concurrent /v1/execute requests run
  data-parallel across the 8 GPUs of a node;
- sandboxes are single-use: taken from the warm deque, torn down after one
  execution, refilled asynchronously off the critical path;
- a freshly spawned sandbox pre-imports numpy and pre-initializes HIP
  (zygote warm child) while it waits in the pool, so those costs are never
  on the request path;
- whole-execution retry x3 with exponential backoff on sandbox failure
  (parity: kubernetes_code_executor.py:75-79).

Also usable as the data plane INSIDE a GPU executor pod, where the pod-level
scheduler is Kubernetes (services/pod_executor.py) and this pool runs with
gpu_count=1.
"""

import asyncio
import logging
import os
import shutil
import signal
import subprocess
import sys
import tempfile
import time
import uuid
from collections import deque
from dataclasses import dataclass
from pathlib import Path
from typing import Mapping, Optional

from code_interpreter_amd.services.code_executor import (
    ExecutorError,
    Result,
    SandboxClient,
)
from code_interpreter_amd.services.storage import Storage
from code_interpreter_amd.utils.gpus import detect_gpu_count
from code_interpreter_amd.utils.retry import async_retry

logger = logging.getLogger("code_executor")

PACKAGE_ROOT = Path(__file__).resolve().parent.parent
RUNTIME_DIR = PACKAGE_ROOT / "executor"
OPS_DIR = PACKAGE_ROOT / "ops"
DEFAULT_SERVER_BIN = RUNTIME_DIR / "build" / "executor-server"


@dataclass
class Sandbox:
    proc: subprocess.Popen
    client: SandboxClient
    root: str
    gpu: Optional[int]

    async def aclose(self) -> None:
        await self.client.aclose()
        try:
            self.proc.send_signal(signal.SIGTERM)
        except ProcessLookupError:
            pass
        try:
            await asyncio.to_thread(self.proc.wait, 5)
        except subprocess.TimeoutExpired:
            self.proc.kill()
            await asyncio.to_thread(self.proc.wait)
        shutil.rmtree(self.root, ignore_errors=True)


class LocalPoolExecutor:
    """Warm pool of single-use local sandboxes, one execution each."""

    def __init__(
        self,
        file_storage: Storage,
        pool_target_length: int = 2,
        gpu_count: int = -1,
        gpu_pinning: bool = True,
        executor_root: str = "",
        server_bin: str = "",
        execute_timeout: float = 60.0,
        zygote_enabled: bool = True,
        scan_recursive: bool = False,
        dep_install: bool = True,
        pip_extra_args: str = "",
        hip_numpy: str = "auto",
        spawn_ready_timeout: float = 60.0,
    ):
        self.file_storage = file_storage
        if gpu_count < 0:
            gpu_count = detect_gpu_count()
        self.gpu_count = gpu_count
        self.gpu_pinning = gpu_pinning and gpu_count > 0
        # pool target scales with GPUs so every device has warm capacity
        self.pool_target_length = pool_target_length * max(1, gpu_count)
        self.executor_root = executor_root or tempfile.mkdtemp(prefix="ci-amd-")
        self.server_bin = server_bin or str(DEFAULT_SERVER_BIN)
        self.execute_timeout = execute_timeout
        self.zygote_enabled = zygote_enabled
        self.scan_recursive = scan_recursive
        self.dep_install = dep_install
        self.pip_extra_args = pip_extra_args
        self.hip_numpy = hip_numpy
        self.spawn_ready_timeout = spawn_ready_timeout

        self._pool: deque[Sandbox] = deque()
        self._spawning_count = 0
        self._next_gpu = 0
        self._closed = False

    # -- lifecycle ---------------------------------------------------------

    async def fill_pool(self) -> None:
        """Top the warm pool up to the target length (async, off the
        request path; parity: fill_executor_pod_queue,
        kubernetes_code_executor.py:151-189)."""
        while (
            not self._closed
            and len(self._pool) + self._spawning_count < self.pool_target_length
        ):
            self._spawning_count += 1
            try:
                sandbox = await self.spawn_sandbox()
                self._pool.append(sandbox)
            except Exception as e:
                logger.warning("sandbox prewarm failed: %s", e)
                await asyncio.sleep(1.0)
                return
            finally:
                self._spawning_count -= 1

    def _pick_gpu(self) -> Optional[int]:
        if not self.gpu_pinning:
            return None
        gpu = self._next_gpu % self.gpu_count
        self._next_gpu += 1
        return gpu

    async def spawn_sandbox(self) -> Sandbox:
        async def attempt() -> Sandbox:
            return await self._spawn_once()

        return await async_retry(attempt, attempts=3, retry_on=(ExecutorError,),
                                 min_backoff=0.5, max_backoff=2.0)

    async def _spawn_once(self) -> Sandbox:
        if not os.path.exists(self.server_bin):
            raise FileNotFoundError(
                f"executor-server binary not found at {self.server_bin}; "
                "build it with `python -m code_interpreter_amd.ops.build` "
                "or `make -C code_interpreter_amd/executor`"
            )
        gpu = self._pick_gpu()
        root = os.path.join(self.executor_root, f"sbx-{uuid.uuid4().hex[:12]}")
        workspace = os.path.join(root, "workspace")
        os.makedirs(workspace, exist_ok=True)
        sock = os.path.join(root, "exec.sock")

        env = dict(os.environ)
        env.update(
            {
                "APP_LISTEN_UNIX": sock,
                "APP_WORKSPACE": workspace,
                "APP_PYTHON": sys.executable,
                "APP_RUNTIME_DIR": str(RUNTIME_DIR),
                "APP_OPS_DIR": str(OPS_DIR),
                "APP_ZYGOTE": "1" if self.zygote_enabled else "0",
                "APP_SCAN_RECURSIVE": "1" if self.scan_recursive else "0",
                "APP_DEP_INSTALL": "1" if self.dep_install else "0",
                "APP_PIP_EXTRA_ARGS": self.pip_extra_args,
                "APP_HIP_NUMPY": self.hip_numpy,
            }
        )
        if gpu is not None:
            env["HIP_VISIBLE_DEVICES"] = str(gpu)

        proc = subprocess.Popen(
            [self.server_bin],
            env=env,
            stdout=subprocess.DEVNULL,
            stderr=subprocess.DEVNULL,
            start_new_session=True,
        )
        client = SandboxClient(uds=sock, timeout=self.execute_timeout + 30.0)
        deadline = time.monotonic() + self.spawn_ready_timeout
        while time.monotonic() < deadline:
            if proc.poll() is not None:
                await client.aclose()
                shutil.rmtree(root, ignore_errors=True)
                raise ExecutorError(
                    f"executor-server exited early (code {proc.returncode})"
                )
            if await client.healthy() is not None:
                return Sandbox(proc=proc, client=client, root=root, gpu=gpu)
            await asyncio.sleep(0.05)
        proc.kill()
        await client.aclose()
        shutil.rmtree(root, ignore_errors=True)
        raise ExecutorError("executor-server did not become ready")

    async def aclose(self) -> None:
        self._closed = True
        while self._pool:
            await self._pool.popleft().aclose()
        shutil.rmtree(self.executor_root, ignore_errors=True)

    # -- execution ---------------------------------------------------------

    async def execute(
        self,
        source_code: str,
        files: Mapping[str, str] = {},
        env: Mapping[str, str] = {},
    ) -> Result:
        async def attempt() -> Result:
            sandbox = await self._take_sandbox()
            try:
                return await sandbox.client.run(
                    self.file_storage,
                    source_code,
                    files=files,
                    env=env,
                    timeout=self.execute_timeout,
                )
            finally:
                # single-use teardown + async refill, off the critical path
                asyncio.create_task(sandbox.aclose())
                asyncio.create_task(self.fill_pool())

        return await async_retry(attempt, attempts=3, retry_on=(ExecutorError,))

    async def _take_sandbox(self) -> Sandbox:
        if self._pool:
            return self._pool.popleft()
        return await self.spawn_sandbox()
