"""Debug helper 2: spawn a REAL service subprocess (python -m
code_interpreter_amd) like bench.py does, run GPU-routed executions over
HTTP, SIGTERM it, and report — used under rocprofv3 to check that the
profiler env survives the extra process layer and the daemon flushes."""

import os
import signal
import socket
import subprocess
import sys
import tempfile
import time
from pathlib import Path

import httpx

REPO = Path(__file__).resolve().parent.parent


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


tmp = tempfile.mkdtemp(prefix="prof-dbg2-")
port = free_port()
env = dict(os.environ)
env.update(
    {
        "APP_HTTP_LISTEN_ADDR": f"127.0.0.1:{port}",
        "APP_GRPC_LISTEN_ADDR": f"127.0.0.1:{free_port()}",
        "APP_FILE_STORAGE_PATH": os.path.join(tmp, "storage"),
        "APP_EXECUTOR_ROOT": os.path.join(tmp, "executors"),
        "APP_EXECUTOR_BACKEND": "local",
        "APP_GPU_COUNT": "1",
        "APP_HIP_NUMPY": "require",
        "APP_DEP_INSTALL": "false",
        "PYTHONPATH": str(REPO),
        "PYTHONFAULTHANDLER": "1",  # SIGABRT dumps all thread stacks
    }
)
err_log = open(os.path.join(tmp, "service.err"), "w")
proc = subprocess.Popen(
    [sys.executable, "-m", "code_interpreter_amd"],
    env=env,
    cwd=str(REPO),
    stdout=err_log,
    stderr=err_log,
    start_new_session=True,
)
code = (
    "import numpy, hipnp\n"
    "x = numpy.random.rand(20_000_000)\n"
    "print(hipnp.backend().name, float(numpy.sum(numpy.square(x))))\n"
)
deadline = time.time() + 40
ok = False
while time.time() < deadline:
    try:
        r = httpx.post(
            f"http://127.0.0.1:{port}/v1/execute",
            json={"source_code": code},
            timeout=40.0,
        )
        if r.status_code == 200 and r.json()["exit_code"] == 0:
            ok = True
            break
    except httpx.HTTPError:
        time.sleep(0.3)
print("first exec ok:", ok, flush=True)
for i in range(3):
    r = httpx.post(
        f"http://127.0.0.1:{port}/v1/execute",
        json={"source_code": code},
        timeout=40.0,
    )
    print("exec", i, r.json()["exit_code"], r.json()["stdout"].strip()[:40])

os.killpg(proc.pid, signal.SIGTERM)
try:
    rc = proc.wait(timeout=20)
    print("service exit code:", rc)
except subprocess.TimeoutExpired:
    print("service HUNG after SIGTERM; dumping thread stacks via SIGABRT")
    os.kill(proc.pid, signal.SIGABRT)
    try:
        proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        os.killpg(proc.pid, signal.SIGKILL)
        proc.wait()
time.sleep(2.0)
err_log.close()
print("---- service stderr tail ----")
print("\n".join(open(os.path.join(tmp, "service.err")).read().splitlines()[-60:]))
