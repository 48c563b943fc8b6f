"""Full-service integration: `python -m code_interpreter_amd` boots both
frontends; the gRPC health check (the k8s liveness probe) passes and HTTP
serves an execution."""

import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import httpx
import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.slow
def test_main_entrypoint_and_health_check(tmp_path, executor_bin):
    http_port = _free_port()
    grpc_port = _free_port()
    env = dict(os.environ)
    env.update(
        {
            "APP_HTTP_LISTEN_ADDR": f"127.0.0.1:{http_port}",
            "APP_GRPC_LISTEN_ADDR": f"127.0.0.1:{grpc_port}",
            "APP_FILE_STORAGE_PATH": str(tmp_path / "storage"),
            "APP_EXECUTOR_ROOT": str(tmp_path / "exec"),
            "APP_GPU_COUNT": "0",
            "APP_EXECUTOR_POOL_TARGET_LENGTH": "1",
            "PYTHONPATH": str(REPO_ROOT),
        }
    )
    proc = subprocess.Popen(
        [sys.executable, "-m", "code_interpreter_amd"],
        env=env,
        cwd=str(REPO_ROOT),
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        start_new_session=True,
    )
    try:
        deadline = time.time() + 30
        up = False
        while time.time() < deadline:
            try:
                r = httpx.post(
                    f"http://127.0.0.1:{http_port}/v1/execute",
                    json={"source_code": "print('boot')"},
                    timeout=30.0,
                )
                if r.status_code == 200 and r.json()["exit_code"] == 0:
                    up = True
                    break
            except httpx.HTTPError:
                time.sleep(0.2)
        assert up, "HTTP frontend did not come up"

        # the k8s liveness probe: gRPC Execute(print(21*2)) == "42\n"
        from code_interpreter_amd.health_check import health_check

        health_check(f"127.0.0.1:{grpc_port}", timeout=60.0)
    finally:
        os.killpg(proc.pid, signal.SIGKILL)
        proc.wait()


@pytest.mark.slow
def test_bench_distributed_contract(tmp_path):
    """The driver launches bench.py under torch.distributed.run for N>1;
    this is the same invocation on CPU (gloo, 2 ranks, tiny array)."""
    import json

    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
            str(REPO_ROOT / "bench.py"),
            "--gpus", "2", "--steps", "4", "--warmup", "1",
            "--array-size", "200000", "--concurrency", "2",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=str(REPO_ROOT),
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = [l for l in proc.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["config"]["parallelism"] == "dp2"
    assert out["steps"] == 4
    assert out["value"] > 0


@pytest.mark.slow
def test_sigterm_graceful_shutdown_reaps_engines(tmp_path, executor_bin):
    """SIGTERM to the service must wind down the whole engine tree
    (executor-server + zygote + warm children) instead of leaking it;
    the service itself exits on its own (no SIGKILL needed)."""
    import psutil

    http_port = _free_port()
    env = dict(os.environ)
    env.update(
        {
            "APP_HTTP_LISTEN_ADDR": f"127.0.0.1:{http_port}",
            "APP_GRPC_LISTEN_ADDR": f"127.0.0.1:{_free_port()}",
            "APP_FILE_STORAGE_PATH": str(tmp_path / "storage"),
            "APP_EXECUTOR_ROOT": str(tmp_path / "exec"),
            "APP_GPU_COUNT": "0",
            "APP_EXECUTOR_POOL_TARGET_LENGTH": "1",
            "PYTHONPATH": str(REPO_ROOT),
        }
    )
    proc = subprocess.Popen(
        [sys.executable, "-m", "code_interpreter_amd"],
        env=env,
        cwd=str(REPO_ROOT),
        stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL,
        start_new_session=True,
    )
    try:
        deadline = time.time() + 30
        up = False
        while time.time() < deadline:
            try:
                r = httpx.post(
                    f"http://127.0.0.1:{http_port}/v1/execute",
                    json={"source_code": "print('x')"},
                    timeout=30.0,
                )
                if r.status_code == 200:
                    up = True
                    break
            except httpx.HTTPError:
                time.sleep(0.2)
        assert up

        service = psutil.Process(proc.pid)
        tree = service.children(recursive=True)
        assert any("executor-server" in p.name() for p in tree), (
            "no engine spawned"
        )

        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=20) == 0

        time.sleep(1.0)  # give reparented grandchildren a beat to exit
        leaked = [p for p in tree if p.is_running()
                  and p.status() != psutil.STATUS_ZOMBIE]
        assert not leaked, f"leaked engine processes: {leaked}"
    finally:
        try:
            os.killpg(proc.pid, signal.SIGKILL)
        except (ProcessLookupError, OSError):
            pass
        proc.wait()
