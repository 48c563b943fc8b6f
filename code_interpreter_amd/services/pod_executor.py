"""Kubernetes executor backend: warm queue of single-use, GPU-pinned
executor pods.

Orchestration parity with the reference's KubernetesCodeExecutor
(kubernetes_code_executor.py:151-264) with the MI355X extensions from
SURVEY.md section 7:

- warm deque topped up to ``pod_queue_target_length`` with in-flight
  spawn tracking; refill runs off the request path;
- pods carry ownerReferences to the service's own pod so Kubernetes
  garbage-collects orphans if the service dies;
- spawn = create pod -> ``kubectl wait --for=condition=Ready``; failures
  delete the half-created pod and retry x3;
- single use: a pod serves exactly one execution, then is deleted
  asynchronously;
- GPU pinning: each pod requests ``amd.com/gpu: 1`` and is assigned
  HIP_VISIBLE_DEVICES round-robin across ``gpu_count`` devices, so
  concurrent /v1/execute requests run data-parallel across the node's
  8 MI355X; arbitrary spec extension stays available through
  ``executor_pod_spec_extra`` / ``executor_container_resources``.

Inside the pod, executor/server.cpp serves the same wire API as locally
(legacy routes, APP_WORKSPACE=/workspace), with its zygote + GPU daemon
giving pre-warmed HIP-ready sandboxes the moment the pod turns Ready.
"""

import asyncio
import collections
import logging
import os
import uuid
from typing import Mapping, Optional

from code_interpreter_amd.services.code_executor import (
    ExecutorError,
    Result,
    SandboxClient,
)
from code_interpreter_amd.services.kubectl import Kubectl
from code_interpreter_amd.services.storage import Storage
from code_interpreter_amd.utils.retry import async_retry

logger = logging.getLogger("code_executor")


class PodExecutor:
    def __init__(
        self,
        kubectl: Kubectl,
        file_storage: Storage,
        executor_image: str,
        container_resources: dict,
        pod_spec_extra: dict,
        pod_queue_target_length: int = 5,
        pod_name_prefix: str = "code-executor-",
        executor_port: int = 8000,
        execute_timeout: float = 60.0,
        gpu_count: int = 0,
        gpu_pinning: bool = True,
        spawn_wait_timeout: str = "60s",
    ):
        self.kubectl = kubectl
        self.file_storage = file_storage
        self.executor_image = executor_image
        self.container_resources = container_resources
        self.pod_spec_extra = pod_spec_extra
        self.pod_queue_target_length = pod_queue_target_length
        self.pod_name_prefix = pod_name_prefix
        self.executor_port = executor_port
        self.execute_timeout = execute_timeout
        self.gpu_count = max(0, gpu_count)
        self.gpu_pinning = gpu_pinning and self.gpu_count > 0
        self.spawn_wait_timeout = spawn_wait_timeout

        self.self_pod: Optional[dict] = None
        self.pod_queue: collections.deque = collections.deque()
        self.spawning_count = 0
        self._next_gpu = 0

    # -- pod lifecycle ----------------------------------------------------

    async def fill_pool(self) -> None:
        """Top the warm pod queue up to target with CONCURRENT in-flight
        spawns (parity: the reference tracks in-flight spawns and runs
        them simultaneously, kubernetes_code_executor.py:151-189). r01
        spawned serially and gave up on the first error; with multi-GB
        ROCm executor images that is minutes of warmup and a fragile
        prefill. Each spawn failure is absorbed and logged (spawn_pod
        already retries x3 internally); the others keep going."""
        need = (
            self.pod_queue_target_length
            - len(self.pod_queue)
            - self.spawning_count
        )
        if need <= 0:
            return
        self.spawning_count += need

        async def spawn_one() -> None:
            try:
                pod = await self.spawn_pod()
                self.pod_queue.append(pod)
            except Exception as e:
                logger.warning("executor pod spawn failed: %s", e)
            finally:
                self.spawning_count -= 1

        await asyncio.gather(*(spawn_one() for _ in range(need)))

    # alias kept for reference-shaped call sites
    fill_executor_pod_queue = fill_pool

    async def _get_self_pod(self) -> Optional[dict]:
        if self.self_pod is None:
            hostname = os.environ.get("HOSTNAME")
            if hostname:
                try:
                    self.self_pod = await self.kubectl.get("pod", hostname)
                except RuntimeError:
                    self.self_pod = None
        return self.self_pod

    def _pick_gpu(self) -> Optional[int]:
        if not self.gpu_pinning:
            return None
        gpu = self._next_gpu % self.gpu_count
        self._next_gpu += 1
        return gpu

    def _pod_manifest(self, name: str, gpu: Optional[int], owner: Optional[dict]) -> dict:
        env = [{"name": "APP_LISTEN_ADDR", "value": f"0.0.0.0:{self.executor_port}"}]
        resources = dict(self.container_resources)
        if gpu is not None:
            env.append({"name": "HIP_VISIBLE_DEVICES", "value": str(gpu)})
            limits = dict(resources.get("limits", {}))
            limits.setdefault("amd.com/gpu", 1)
            resources["limits"] = limits
        manifest = {
            "apiVersion": "v1",
            "kind": "Pod",
            "metadata": {
                "name": name,
                "labels": {"app": "code-interpreter-amd-executor"},
                **(
                    {
                        "ownerReferences": [
                            {
                                "apiVersion": "v1",
                                "kind": "Pod",
                                "name": owner["metadata"]["name"],
                                "uid": owner["metadata"]["uid"],
                            }
                        ]
                    }
                    if owner
                    else {}
                ),
            },
            "spec": {
                "restartPolicy": "Never",
                "containers": [
                    {
                        "name": "executor",
                        "image": self.executor_image,
                        "ports": [{"containerPort": self.executor_port}],
                        "env": env,
                        "resources": resources,
                    }
                ],
                **self.pod_spec_extra,
            },
        }
        return manifest

    async def spawn_pod(self) -> dict:
        async def attempt() -> dict:
            return await self._spawn_pod_once()

        return await async_retry(attempt, attempts=3, retry_on=(RuntimeError,))

    async def _spawn_pod_once(self) -> dict:
        name = f"{self.pod_name_prefix}{uuid.uuid4().hex[:10]}"
        owner = await self._get_self_pod()
        gpu = self._pick_gpu()
        manifest = self._pod_manifest(name, gpu, owner)
        await self.kubectl.create(body=manifest)
        try:
            await self.kubectl.wait(
                f"pod/{name}",
                **{"for": "condition=Ready", "timeout": self.spawn_wait_timeout},
            )
            pod = await self.kubectl.get("pod", name)
            logger.info("executor pod %s ready (gpu=%s)", name, gpu)
            return pod
        except RuntimeError:
            # best-effort cleanup of the half-created pod, then retry
            try:
                await self.kubectl.delete("pod", name)
            except RuntimeError:
                pass
            raise

    async def _take_pod(self) -> dict:
        if self.pod_queue:
            pod = self.pod_queue.popleft()
        else:
            pod = await self.spawn_pod()
        # async refill, off the critical path
        asyncio.ensure_future(self.fill_pool())
        return pod

    async def _delete_pod(self, pod: dict) -> None:
        try:
            await self.kubectl.delete("pod", pod["metadata"]["name"])
        except RuntimeError as e:
            logger.warning("pod delete failed: %s", e)

    # -- execution --------------------------------------------------------

    async def execute(
        self,
        source_code: str,
        files: Mapping[str, str] = {},
        env: Mapping[str, str] = {},
    ) -> Result:
        async def attempt() -> Result:
            pod = await self._take_pod()
            pod_ip = pod["status"]["podIP"]
            client = SandboxClient(
                base_url=f"http://{pod_ip}:{self.executor_port}",
                timeout=self.execute_timeout + 30.0,
            )
            try:
                return await client.run(
                    self.file_storage,
                    source_code,
                    files=files,
                    env=env,
                    timeout=self.execute_timeout,
                )
            finally:
                await client.aclose()
                # single-use pod teardown, async (parity: :262-264)
                asyncio.ensure_future(self._delete_pod(pod))

        return await async_retry(
            attempt, attempts=3, retry_on=(ExecutorError, RuntimeError)
        )

    async def aclose(self) -> None:
        while self.pod_queue:
            await self._delete_pod(self.pod_queue.popleft())
