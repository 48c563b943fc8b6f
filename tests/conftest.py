import asyncio
import socket
import subprocess
import sys
import threading
import time
from pathlib import Path

import httpx
import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))

EXECUTOR_BIN = (
    REPO_ROOT / "code_interpreter_amd" / "executor" / "build" / "executor-server"
)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on GPU box)")
    config.addinivalue_line("markers", "slow: slower integration tests")


def _ensure_executor_built():
    if not EXECUTOR_BIN.exists():
        subprocess.run(
            ["make", "-C", str(EXECUTOR_BIN.parent.parent)],
            check=True,
            capture_output=True,
        )


@pytest.fixture(scope="session")
def executor_bin():
    _ensure_executor_built()
    return str(EXECUTOR_BIN)


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class ServiceUnderTest:
    """The full service (FastAPI + local executor pool) on a real uvicorn
    server in a background thread -- the tests drive it exactly the way the
    reference e2e suite drives a port-forwarded deployment."""

    def __init__(self, tmp_path: Path, gpu_count: int = 0, **config_overrides):
        from code_interpreter_amd.application_context import ApplicationContext
        from code_interpreter_amd.config import Config

        self.port = _free_port()
        self.config = Config(
            http_listen_addr=f"127.0.0.1:{self.port}",
            file_storage_path=str(tmp_path / "storage"),
            executor_root=str(tmp_path / "executors"),
            executor_backend="local",
            executor_pool_target_length=1,
            gpu_count=gpu_count,
            **config_overrides,
        )
        self.ctx = ApplicationContext(self.config)
        self._thread = None
        self._server = None

    def start(self):
        import uvicorn

        app = self.ctx.http_server
        self._server = uvicorn.Server(
            uvicorn.Config(
                app=app, host="127.0.0.1", port=self.port, log_config=None
            )
        )
        self._thread = threading.Thread(target=self._server.run, daemon=True)
        self._thread.start()
        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                httpx.get(f"http://127.0.0.1:{self.port}/docs", timeout=1.0)
                return self
            except httpx.HTTPError:
                time.sleep(0.05)
        raise RuntimeError("service did not start")

    def stop(self):
        if self._server:
            self._server.should_exit = True
        if self._thread:
            self._thread.join(timeout=10)
        asyncio.run(self.ctx.code_executor.aclose())

    @property
    def base_url(self) -> str:
        return f"http://127.0.0.1:{self.port}"


@pytest.fixture(scope="session")
def service(tmp_path_factory, executor_bin):
    svc = ServiceUnderTest(tmp_path_factory.mktemp("svc")).start()
    yield svc
    svc.stop()


@pytest.fixture(scope="session")
def http_client(service):
    with httpx.Client(base_url=service.base_url, timeout=120.0) as client:
        yield client


@pytest.fixture(scope="session")
def wheelhouse(tmp_path_factory):
    """A local wheelhouse with a tiny 'cowsay'-style package, so the
    on-the-fly dependency-install path is testable without a network
    (pip gets --no-index --find-links <wheelhouse>)."""
    root = tmp_path_factory.mktemp("wheelhouse")
    pkg = root / "src" / "mootool"
    pkg.mkdir(parents=True)
    (pkg / "__init__.py").write_text(
        'def moo(text):\n    print("moo says: " + text)\n'
    )
    (root / "src" / "setup.py").write_text(
        "from setuptools import setup\n"
        'setup(name="mootool", version="1.0", packages=["mootool"])\n'
    )
    subprocess.run(
        [
            sys.executable,
            "-m",
            "pip",
            "wheel",
            "--no-deps",
            "--no-build-isolation",
            "--no-index",
            "-w",
            str(root),
            str(root / "src"),
        ],
        check=True,
        capture_output=True,
    )
    return str(root)


def run_async(coro):
    return asyncio.run(coro)


@pytest.fixture(scope="session")
def gpu_executor(tmp_path_factory, executor_bin):
    """A live 1-GPU local executor pool (skipped when no GPU is visible)."""
    from code_interpreter_amd.services.local_executor import LocalPoolExecutor
    from code_interpreter_amd.services.storage import Storage
    from code_interpreter_amd.utils.gpus import detect_gpu_count

    # sysfs probe, NOT _hipops.is_available(): initializing the HIP
    # runtime in the pytest process before torch makes torch.cuda blind
    # (NOTES.md binding rules) -- this fixture must stay HIP-free
    if detect_gpu_count() == 0:
        pytest.skip("no AMD GPU visible")

    tmp = tmp_path_factory.mktemp("gpue")
    ex = LocalPoolExecutor(
        Storage(str(tmp / "storage")),
        pool_target_length=2,
        gpu_count=1,
        executor_root=str(tmp / "eng"),
        hip_numpy="require",
        dep_install=False,
        execute_timeout=120.0,
    )
    yield ex
    asyncio.run(ex.aclose())
