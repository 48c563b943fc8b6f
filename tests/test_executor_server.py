"""Direct tests of the C++ executor-server data plane (executor/server.cpp):
workspace upload/download, execution semantics, timeout behavior, and the
changed-file scan (reference parity: server.rs:69-179)."""

import asyncio
import os
import subprocess
import sys
import time
from pathlib import Path

import httpx
import pytest

RUNTIME_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "executor"


class RawExecutor:
    def __init__(self, tmp_path: Path, executor_bin: str, **env_overrides):
        self.root = tmp_path
        self.workspace = tmp_path / "workspace"
        self.workspace.mkdir(exist_ok=True)
        self.sock = str(tmp_path / "x.sock")
        env = dict(os.environ)
        env.update(
            {
                "APP_LISTEN_UNIX": self.sock,
                "APP_WORKSPACE": str(self.workspace),
                "APP_PYTHON": sys.executable,
                "APP_RUNTIME_DIR": str(RUNTIME_DIR),
                "APP_DEP_INSTALL": "0",
                "APP_HIP_NUMPY": "off",
            }
        )
        env.update(env_overrides)
        self.proc = subprocess.Popen(
            [executor_bin], env=env, start_new_session=True,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        )
        transport = httpx.HTTPTransport(uds=self.sock)
        self.client = httpx.Client(
            base_url="http://x", transport=transport, timeout=60.0
        )
        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                if self.client.get("/healthz").status_code == 200:
                    return
            except httpx.HTTPError:
                time.sleep(0.05)
        raise RuntimeError("executor did not start")

    def close(self):
        self.client.close()
        self.proc.kill()
        self.proc.wait()


@pytest.fixture
def raw_executor(tmp_path, executor_bin):
    ex = RawExecutor(tmp_path, executor_bin)
    yield ex
    ex.close()


def test_upload_download_roundtrip(raw_executor):
    data = b"x" * 100_000 + "ünïcode".encode()
    resp = raw_executor.client.put("/workspace/sub/dir/blob.bin", content=data)
    assert resp.status_code == 204
    assert (raw_executor.workspace / "sub" / "dir" / "blob.bin").read_bytes() == data
    resp = raw_executor.client.get("/workspace/sub/dir/blob.bin")
    assert resp.status_code == 200
    assert resp.content == data


def test_chunked_upload(raw_executor):
    def gen():
        for _ in range(10):
            yield b"chunk" * 1000

    resp = raw_executor.client.put("/workspace/streamed.bin", content=gen())
    assert resp.status_code == 204
    assert (raw_executor.workspace / "streamed.bin").stat().st_size == 50_000


def test_download_missing_404(raw_executor):
    assert raw_executor.client.get("/workspace/nope.txt").status_code == 404


def test_path_traversal_rejected(raw_executor):
    # %2e%2e decodes to ".." server-side
    assert (
        raw_executor.client.put(
            "/workspace/%2e%2e/evil2.txt", content=b"x"
        ).status_code
        == 400
    )
    # raw socket: httpx would normalize a literal "..", the server must too
    import socket as socketmod

    s = socketmod.socket(socketmod.AF_UNIX, socketmod.SOCK_STREAM)
    s.connect(raw_executor.sock)
    s.sendall(
        b"PUT /workspace/../evil.txt HTTP/1.1\r\nHost: x\r\n"
        b"Content-Length: 1\r\nConnection: close\r\n\r\nx"
    )
    status_line = b""
    while b"\r\n" not in status_line:
        chunk = s.recv(4096)
        if not chunk:
            break
        status_line += chunk
    s.close()
    assert b" 400 " in status_line.split(b"\r\n")[0]


def test_execute_basic(raw_executor):
    resp = raw_executor.client.post(
        "/execute", json={"source_code": "print('hi')"}
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["stdout"] == "hi\n"
    assert body["exit_code"] == 0
    assert body["files"] == []


def test_execute_timeout(raw_executor):
    t0 = time.time()
    resp = raw_executor.client.post(
        "/execute",
        json={"source_code": "import time\ntime.sleep(60)", "timeout": 2},
    )
    elapsed = time.time() - t0
    assert elapsed < 20
    body = resp.json()
    # exact reference behavior: ("", "Execution timed out", -1), server.rs:169
    assert body["stdout"] == ""
    assert body["stderr"] == "Execution timed out"
    assert body["exit_code"] == -1


def test_changed_file_scan_nonrecursive_parity(raw_executor):
    """Reference parity: only top-level /workspace files are reported
    (server.rs:98-118 is non-recursive); nested outputs are not returned."""
    resp = raw_executor.client.post(
        "/execute",
        json={
            "source_code": (
                "import os\n"
                "open('top.txt','w').write('t')\n"
                "os.makedirs('nested', exist_ok=True)\n"
                "open('nested/inner.txt','w').write('i')\n"
            )
        },
    )
    files = resp.json()["files"]
    assert files == ["/workspace/top.txt"]


def test_changed_file_scan_recursive_flag(tmp_path, executor_bin):
    ex = RawExecutor(tmp_path, executor_bin, APP_SCAN_RECURSIVE="1")
    try:
        resp = ex.client.post(
            "/execute",
            json={
                "source_code": (
                    "import os\n"
                    "open('top.txt','w').write('t')\n"
                    "os.makedirs('nested', exist_ok=True)\n"
                    "open('nested/inner.txt','w').write('i')\n"
                )
            },
        )
        assert sorted(resp.json()["files"]) == [
            "/workspace/nested/inner.txt",
            "/workspace/top.txt",
        ]
    finally:
        ex.close()


def test_preexisting_inputs_not_reported(raw_executor):
    """Files uploaded before execution start must not appear as changed
    unless the code rewrites them."""
    raw_executor.client.put("/workspace/input.txt", content=b"in")
    time.sleep(0.05)
    resp = raw_executor.client.post(
        "/execute", json={"source_code": "print(open('input.txt').read())"}
    )
    body = resp.json()
    assert body["stdout"] == "in\n"
    assert body["files"] == []


def test_env_merged_not_replaced(raw_executor):
    resp = raw_executor.client.post(
        "/execute",
        json={
            "source_code": "import os\nprint(os.environ['A'], len(os.environ.get('PATH','')) > 0)",
            "env": {"A": "1"},
        },
    )
    assert resp.json()["stdout"] == "1 True\n"


def test_concurrent_executes(raw_executor):
    """The data plane must handle parallel requests (thread-per-conn)."""

    async def run_all():
        transport = httpx.AsyncHTTPTransport(uds=raw_executor.sock)
        async with httpx.AsyncClient(
            base_url="http://x", transport=transport, timeout=60.0
        ) as client:
            async def one(i):
                r = await client.post(
                    "/execute", json={"source_code": f"print({i}*{i})"}
                )
                return r.json()["stdout"].strip()

            return await asyncio.gather(*(one(i) for i in range(8)))

    results = asyncio.run(run_all())
    assert results == [str(i * i) for i in range(8)]


def test_cold_mode_no_zygote(tmp_path, executor_bin):
    ex = RawExecutor(tmp_path, executor_bin, APP_ZYGOTE="0")
    try:
        resp = ex.client.post("/execute", json={"source_code": "print('cold')"})
        assert resp.json()["stdout"] == "cold\n"
        assert resp.json()["exit_code"] == 0
    finally:
        ex.close()


# ---------------------------------------------------------------------------
# `!cmd` shell escapes (reference: xonsh, server.rs:152-165; here a
# compile-gated source transform in sandbox_runtime)
# ---------------------------------------------------------------------------
SHELL_ESCAPE_SCRIPT = """x = 6 * 7
!echo "shell says $x"
print("python says", x)
if x > 10:
    !echo nested-escape-ran
!false
print("still alive after failing command")
"""


def test_shell_escape_lines(raw_executor):
    r = raw_executor.client.post(
        "/execute", json={"source_code": SHELL_ESCAPE_SCRIPT}
    )
    body = r.json()
    assert body["exit_code"] == 0, body["stderr"]
    out = body["stdout"]
    # $x is SHELL interpolation: undefined in the shell env -> empty
    assert "shell says" in out
    assert "python says 42" in out
    assert "nested-escape-ran" in out
    assert "still alive after failing command" in out


def test_shell_escape_cold_path(tmp_path, executor_bin):
    # APP_ZYGOTE=0: fork/exec fallback must run the same sandbox runtime
    ex = RawExecutor(tmp_path, executor_bin, APP_ZYGOTE="0")
    try:
        r = ex.client.post(
            "/execute", json={"source_code": '!echo cold-escape-ok\nprint("py")'}
        )
        body = r.json()
        assert body["exit_code"] == 0, body["stderr"]
        assert "cold-escape-ok" in body["stdout"]
        assert "py" in body["stdout"]
    finally:
        ex.close()


def test_plain_python_with_bang_in_string_untouched(raw_executor):
    # a multiline string containing a `!`-leading line is valid python:
    # the transform must never fire
    src = 's = """\n!not a command\n"""\nprint(repr(s))'
    r = raw_executor.client.post("/execute", json={"source_code": src})
    body = r.json()
    assert body["exit_code"] == 0
    assert "!not a command" in body["stdout"]


def test_real_syntax_error_still_reported(raw_executor):
    r = raw_executor.client.post(
        "/execute", json={"source_code": "def broken(:\n    pass"}
    )
    body = r.json()
    assert body["exit_code"] == 1
    assert "SyntaxError" in body["stderr"]


def test_shell_capture_expression(raw_executor):
    src = (
        'listing = $(echo alpha beta)\n'
        'print("captured:", listing.strip())\n'
        'n = len($(echo one two three).split())\n'
        'print("words:", n)\n'
    )
    r = raw_executor.client.post("/execute", json={"source_code": src})
    body = r.json()
    assert body["exit_code"] == 0, body["stderr"]
    assert "captured: alpha beta" in body["stdout"]
    assert "words: 3" in body["stdout"]
