"""Execute scripts/run-local.sh for real (the dev scripts
had never run): it must build the native components, boot the service,
and answer /v1/execute end to end."""

import os
import signal
import socket
import subprocess
import time
from pathlib import Path

import httpx
import pytest

REPO = Path(__file__).resolve().parent.parent


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(300)
def test_run_local_sh_boots_and_serves(tmp_path):
    http_port = _free_port()
    env = dict(os.environ)
    env.update(
        {
            "APP_HTTP_LISTEN_ADDR": f"127.0.0.1:{http_port}",
            "APP_GRPC_LISTEN_ADDR": f"127.0.0.1:{_free_port()}",
            "APP_FILE_STORAGE_PATH": str(tmp_path / "storage"),
            "APP_EXECUTOR_ROOT": str(tmp_path / "executors"),
            "APP_EXECUTOR_BACKEND": "local",
            "APP_EXECUTOR_POOL_TARGET_LENGTH": "1",
            "APP_ENGINES_PER_GPU": "1",
            "APP_GPU_COUNT": "0",
            "APP_HIP_NUMPY": "off",
            "APP_DEP_INSTALL": "false",
        }
    )
    proc = subprocess.Popen(
        ["bash", str(REPO / "scripts" / "run-local.sh")],
        env=env,
        cwd=str(REPO),
        stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL,
        start_new_session=True,
    )
    try:
        with httpx.Client(
            base_url=f"http://127.0.0.1:{http_port}", timeout=10.0
        ) as client:
            deadline = time.time() + 240
            body = None
            while time.time() < deadline:
                if proc.poll() is not None:
                    pytest.fail(f"run-local.sh exited early rc={proc.returncode}")
                try:
                    r = client.post(
                        "/v1/execute",
                        json={"source_code": "print(6 * 7)"},
                    )
                    if r.status_code == 200:
                        body = r.json()
                        break
                except httpx.HTTPError:
                    pass
                time.sleep(0.5)
            assert body is not None, "service never became ready"
            assert body["exit_code"] == 0
            assert body["stdout"] == "42\n"
    finally:
        try:
            os.killpg(proc.pid, signal.SIGTERM)
        except (ProcessLookupError, OSError):
            proc.terminate()
        try:
            proc.wait(20)
        except subprocess.TimeoutExpired:
            os.killpg(proc.pid, signal.SIGKILL)
            proc.wait()
