"""Shared executor-side types and the client for the in-sandbox HTTP API.

The internal sandbox API (served by executor/server.cpp):

  legacy pod-style (wire-compatible with the reference's in-pod API,
  server.rs:189-191; used by the kubernetes pod backend where the pod
  itself is single-use):
    PUT/GET /workspace/{rel}         file staging
    POST    /execute                 {source_code, timeout?, env?} ->
                                     {stdout, stderr, exit_code, files:[...]}

  engine-style sessions (used by the local per-GPU engine backend; the
  single-use unit is a forked interpreter + fresh workspace, not the
  server):
    POST    /sandboxes               -> {id}
    PUT/GET /sandboxes/{id}/workspace/{rel}
    POST    /sandboxes/{id}/execute
    DELETE  /sandboxes/{id}

Both backends drive a sandbox through SandboxClient.run(): parallel-upload
input files from storage, execute, parallel-download changed files back
into storage (reference orchestration shape:
kubernetes_code_executor.py:81-149).
"""

import asyncio
from contextvars import ContextVar
from dataclasses import dataclass, field
from typing import Mapping, Optional

import httpx

# request-id propagation into the executor (logged there; SURVEY.md
# section 5: pass the request id to the pod as a header)
REQUEST_ID: ContextVar = ContextVar("sandbox_request_id", default="-")

from code_interpreter_amd.services.storage import Storage

WORKSPACE_PREFIX = "/workspace/"


@dataclass
class Result:
    stdout: str
    stderr: str
    exit_code: int
    files: Mapping[str, str] = field(default_factory=dict)
    # per-stage timings (ms): sandbox create/upload/exec/download on the
    # client side plus dispatch/run from the executor server
    timings: Mapping[str, float] = field(default_factory=dict)


class ExecutorError(RuntimeError):
    """Sandbox-level failure (unreachable, bad response, ...). Retryable."""


def _rel(path: str) -> str:
    if path.startswith(WORKSPACE_PREFIX):
        return path[len(WORKSPACE_PREFIX):]
    return path.lstrip("/")


class SandboxClient:
    """HTTP client for one executor server (TCP base_url or unix socket)."""

    def __init__(
        self,
        base_url: str = "http://executor",
        uds: Optional[str] = None,
        timeout: float = 60.0,
    ):
        transport = (
            httpx.AsyncHTTPTransport(uds=uds, limits=httpx.Limits(max_connections=64))
            if uds
            else None
        )
        self._client = httpx.AsyncClient(
            base_url=base_url, transport=transport, timeout=timeout
        )

    async def aclose(self) -> None:
        await self._client.aclose()

    async def healthy(self) -> Optional[dict]:
        try:
            resp = await self._client.get("/healthz", timeout=2.0)
            if resp.status_code == 200:
                return resp.json()
        except (httpx.HTTPError, OSError):
            pass
        return None

    async def create_sandbox(self) -> str:
        resp = await self._client.post("/sandboxes")
        if resp.status_code != 200:
            raise ExecutorError(f"sandbox create failed: {resp.status_code}")
        return resp.json()["id"]

    async def delete_sandbox(self, sandbox_id: str) -> None:
        try:
            await self._client.delete(f"/sandboxes/{sandbox_id}")
        except (httpx.HTTPError, OSError, RuntimeError):
            pass  # best-effort teardown (client may already be closed)

    async def run(
        self,
        storage: Storage,
        source_code: str,
        files: Mapping[str, str] = {},
        env: Mapping[str, str] = {},
        timeout: Optional[float] = None,
        session: Optional[str] = None,
    ) -> Result:
        """Run one execution. With ``session``, all routes are scoped to
        that sandbox session's fresh workspace."""
        prefix = f"/sandboxes/{session}" if session else ""
        try:
            return await self._run(storage, source_code, files, env, timeout, prefix)
        except (httpx.HTTPError, OSError) as e:
            raise ExecutorError(f"sandbox request failed: {e!r}") from e

    async def run_single_use(
        self,
        storage: Storage,
        source_code: str,
        files: Mapping[str, str] = {},
        env: Mapping[str, str] = {},
        timeout: Optional[float] = None,
    ) -> Result:
        """Run in a fresh single-use sandbox session. Without input files
        the whole round is ONE request (/execute-ephemeral: the engine
        creates the workspace, runs, and deletes it when no files changed);
        with input files (or changed outputs) the explicit session routes
        handle staging, then the session is torn down."""
        import time as _time

        if not files:
            try:
                return await self._run_ephemeral(storage, source_code, env, timeout)
            except (httpx.HTTPError, OSError) as e:
                raise ExecutorError(f"sandbox request failed: {e!r}") from e

        t0 = _time.perf_counter()
        try:
            session = await self.create_sandbox()
        except (httpx.HTTPError, OSError) as e:
            raise ExecutorError(f"sandbox create failed: {e!r}") from e
        t_create = (_time.perf_counter() - t0) * 1000
        try:
            result = await self.run(
                storage, source_code, files=files, env=env, timeout=timeout,
                session=session,
            )
            result.timings = {**result.timings, "create_ms": round(t_create, 2)}
            return result
        finally:
            asyncio.ensure_future(self.delete_sandbox(session))

    async def _run_ephemeral(self, storage, source_code, env, timeout) -> Result:
        body: dict = {"source_code": source_code, "env": dict(env)}
        if timeout is not None:
            body["timeout"] = timeout
        headers = {}
        request_id = REQUEST_ID.get()
        if request_id and request_id != "-":
            headers["X-Request-Id"] = request_id
        import time as _time

        t0 = _time.perf_counter()
        resp = await self._client.post(
            "/execute-ephemeral", json=body, headers=headers
        )
        t_exec = (_time.perf_counter() - t0) * 1000
        if resp.status_code != 200:
            raise ExecutorError(f"execute failed: {resp.status_code} {resp.text!r}")
        payload = resp.json()
        session = payload.get("session")
        stored = {}
        if payload["files"] and session:
            try:
                prefix = f"/sandboxes/{session}"

                async def download(path: str):
                    async with storage.writer() as writer:
                        async with self._client.stream(
                            "GET", f"{prefix}/workspace/{_rel(path)}"
                        ) as file_resp:
                            file_resp.raise_for_status()
                            async for chunk in file_resp.aiter_bytes():
                                await writer.write(chunk)
                        return path, writer.hash

                stored = dict(
                    await asyncio.gather(*(download(p) for p in payload["files"]))
                )
            finally:
                asyncio.ensure_future(self.delete_sandbox(session))
        timings = dict(payload.get("timings") or {})
        timings["exec_api_ms"] = round(t_exec, 2)
        return Result(
            stdout=payload["stdout"],
            stderr=payload["stderr"],
            exit_code=payload["exit_code"],
            files=stored,
            timings=timings,
        )

    async def _run(self, storage, source_code, files, env, timeout, prefix) -> Result:
        async def upload(path: str, object_hash: str):
            async with storage.reader(object_hash) as reader:
                data = await reader.read()
            resp = await self._client.put(
                f"{prefix}/workspace/{_rel(path)}", content=data
            )
            if resp.status_code not in (200, 204):
                raise ExecutorError(f"upload of {path} failed: {resp.status_code}")

        import time as _time

        t0 = _time.perf_counter()
        await asyncio.gather(*(upload(p, h) for p, h in files.items()))
        t_upload = (_time.perf_counter() - t0) * 1000

        body: dict = {"source_code": source_code, "env": dict(env)}
        if timeout is not None:
            body["timeout"] = timeout
        headers = {}
        request_id = REQUEST_ID.get()
        if request_id and request_id != "-":
            headers["X-Request-Id"] = request_id
        t0 = _time.perf_counter()
        resp = await self._client.post(
            f"{prefix}/execute", json=body, headers=headers
        )
        t_exec = (_time.perf_counter() - t0) * 1000
        if resp.status_code != 200:
            raise ExecutorError(f"execute failed: {resp.status_code} {resp.text!r}")
        payload = resp.json()

        async def download(path: str):
            async with storage.writer() as writer:
                async with self._client.stream(
                    "GET", f"{prefix}/workspace/{_rel(path)}"
                ) as file_resp:
                    file_resp.raise_for_status()
                    async for chunk in file_resp.aiter_bytes():
                        await writer.write(chunk)
                return path, writer.hash

        t0 = _time.perf_counter()
        stored = dict(await asyncio.gather(*(download(p) for p in payload["files"])))
        t_download = (_time.perf_counter() - t0) * 1000
        timings = dict(payload.get("timings") or {})
        timings.update(
            upload_ms=round(t_upload, 2),
            exec_api_ms=round(t_exec, 2),
            download_ms=round(t_download, 2),
        )
        return Result(
            stdout=payload["stdout"],
            stderr=payload["stderr"],
            exit_code=payload["exit_code"],
            files=stored,
            timings=timings,
        )
