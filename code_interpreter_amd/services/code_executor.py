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

import aiohttp

# everything a sandbox HTTP call can raise that should read as "sandbox
# unreachable / failed" (retryable at the executor layer)
CLIENT_ERRORS = (aiohttp.ClientError, asyncio.TimeoutError, OSError)

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
    """HTTP client for one executor server (TCP base_url or unix socket).
    aiohttp rather than httpx: measured ~3 vs ~1 CPU-ms per request
    client-side, and the serving path is bounded by the container CPU
    quota (profiles/NOTES.md)."""

    def __init__(
        self,
        base_url: str = "http://executor",
        uds: Optional[str] = None,
        timeout: float = 60.0,
    ):
        self._uds = uds
        self._base_url = base_url if not uds else "http://executor"
        self._timeout = aiohttp.ClientTimeout(total=timeout)
        self._session: Optional[aiohttp.ClientSession] = None
        self._loop = None

    @staticmethod
    def _abandon(session: aiohttp.ClientSession) -> None:
        """Free a session bound to a finished/foreign event loop without
        awaiting on it (aiohttp sessions are loop-affine; test harnesses
        run each call in its own asyncio.run loop). Closes the raw
        sockets directly; the session object is then dropped."""
        try:
            conn = session._connector
            if conn is not None:
                for dq in list(getattr(conn, "_conns", {}).values()):
                    for item in list(dq):
                        proto = item[0] if isinstance(item, (tuple, list)) else item
                        tr = getattr(proto, "transport", None)
                        if tr is None:
                            continue
                        sock = tr.get_extra_info("socket")
                        if sock is not None:
                            try:
                                import warnings

                                with warnings.catch_warnings():
                                    warnings.simplefilter("ignore")
                                    sock.close()
                            except OSError:
                                pass
                conn._closed = True
            session._connector = None
        except Exception:
            pass

    def _client(self) -> aiohttp.ClientSession:
        # lazy + per-loop: a ClientSession must be created AND used on one
        # running event loop
        loop = asyncio.get_running_loop()
        if self._session is None or self._session.closed or self._loop is not loop:
            if self._session is not None and not self._session.closed:
                self._abandon(self._session)
            connector = (
                aiohttp.UnixConnector(path=self._uds, limit=64)
                if self._uds
                else aiohttp.TCPConnector(limit=64)
            )
            self._session = aiohttp.ClientSession(
                base_url=self._base_url,
                connector=connector,
                timeout=self._timeout,
            )
            self._loop = loop
        return self._session

    async def aclose(self) -> None:
        session, self._session = self._session, None
        if session is None or session.closed:
            return
        try:
            if self._loop is asyncio.get_running_loop() and not self._loop.is_closed():
                await session.close()
                return
        except RuntimeError:
            pass
        self._abandon(session)

    async def healthy(self) -> Optional[dict]:
        try:
            async with self._client().get(
                "/healthz", timeout=aiohttp.ClientTimeout(total=2.0)
            ) as resp:
                if resp.status == 200:
                    return await resp.json()
        except CLIENT_ERRORS:
            pass
        return None

    async def create_sandbox(self) -> str:
        async with self._client().post("/sandboxes") as resp:
            if resp.status != 200:
                raise ExecutorError(f"sandbox create failed: {resp.status}")
            return (await resp.json())["id"]

    async def delete_sandbox(self, sandbox_id: str) -> None:
        try:
            async with self._client().delete(f"/sandboxes/{sandbox_id}"):
                pass
        except CLIENT_ERRORS + (RuntimeError,):
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
        except CLIENT_ERRORS as e:
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
            except CLIENT_ERRORS as e:
                raise ExecutorError(f"sandbox request failed: {e!r}") from e

        t0 = _time.perf_counter()
        try:
            session = await self.create_sandbox()
        except CLIENT_ERRORS as e:
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
        async with self._client().post(
            "/execute-ephemeral", json=body, headers=headers
        ) as resp:
            t_exec = (_time.perf_counter() - t0) * 1000
            if resp.status != 200:
                text = await resp.text()
                raise ExecutorError(f"execute failed: {resp.status} {text!r}")
            payload = await resp.json()
        session = payload.get("session")
        stored = {}
        if payload["files"] and session:
            try:
                prefix = f"/sandboxes/{session}"

                async def download(path: str):
                    async with storage.writer() as writer:
                        async with self._client().get(
                            f"{prefix}/workspace/{_rel(path)}"
                        ) as file_resp:
                            file_resp.raise_for_status()
                            async for chunk in file_resp.content.iter_chunked(1 << 16):
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
            # stream in 64 KiB chunks (chunked transfer into the
            # executor's streaming PUT): a multi-GB input file must not
            # be buffered whole in the control plane
            async def chunks():
                async with storage.reader(object_hash) as reader:
                    while True:
                        piece = await reader.read(1 << 16)
                        if not piece:
                            break
                        yield piece

            async with self._client().put(
                f"{prefix}/workspace/{_rel(path)}", data=chunks()
            ) as resp:
                if resp.status not in (200, 204):
                    raise ExecutorError(f"upload of {path} failed: {resp.status}")

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
        async with self._client().post(
            f"{prefix}/execute", json=body, headers=headers
        ) as resp:
            t_exec = (_time.perf_counter() - t0) * 1000
            if resp.status != 200:
                text = await resp.text()
                raise ExecutorError(f"execute failed: {resp.status} {text!r}")
            payload = await resp.json()

        async def download(path: str):
            async with storage.writer() as writer:
                async with self._client().get(
                    f"{prefix}/workspace/{_rel(path)}"
                ) as file_resp:
                    file_resp.raise_for_status()
                    async for chunk in file_resp.content.iter_chunked(1 << 16):
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
