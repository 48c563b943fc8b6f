"""CPU rehearsal of the driver's multi-GPU bench invocation (VERDICT r01
item 3): the exact `python -m torch.distributed.run --nnodes=1
--nproc-per-node 8 bench.py --gpus 8` path — rank env parsing, per-rank
service spawn, barriers, max-elapsed aggregation, rank-0 single-JSON-line
contract — exercised with the gloo backend and a tiny workload so the
driver's first real 8-GPU SCALE run has no untested code on its path.
"""

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent

sys.path.insert(0, str(REPO_ROOT))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_rank_visible_device_reslicing():
    import bench

    # bare environment: rank n -> device n
    assert bench._rank_visible_device(None, 0) == "0"
    assert bench._rank_visible_device("", 3) == "3"
    # pre-restricted list: each rank takes ITS entry, not the whole list
    assert bench._rank_visible_device("4,5,6,7", 0) == "4"
    assert bench._rank_visible_device("4,5,6,7", 3) == "7"
    assert bench._rank_visible_device("2", 0) == "2"
    # ragged list with blanks
    assert bench._rank_visible_device("1,,3", 1) == "3"


@pytest.mark.timeout(540)
def test_torchrun_8rank_gloo_rehearsal(tmp_path):
    env = dict(os.environ)
    env.update(
        {
            "APP_BENCH_WARM_MAX_S": "3",  # tiny workload: no long warm loop
            "MASTER_ADDR": "127.0.0.1",
            "PYTHONPATH": str(REPO_ROOT),
        }
    )
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "8",
        "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
        str(REPO_ROOT / "bench.py"),
        "--gpus", "8", "--steps", "8", "--warmup", "1",
        "--concurrency", "2", "--array-size", "10000",
    ]
    proc = subprocess.run(
        cmd, env=env, cwd=str(REPO_ROOT), capture_output=True, text=True,
        timeout=480,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    json_lines = [
        line for line in proc.stdout.splitlines()
        if line.startswith("{") and '"metric"' in line
    ]
    # rank 0 prints exactly ONE JSON line for the whole job
    assert len(json_lines) == 1, proc.stdout[-2000:]
    rec = json.loads(json_lines[0])
    assert rec["steps"] == 8
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp8"
    assert rec["n_gpus"] == 0  # CPU rehearsal: no GPU claimed
    assert rec["p50_ms"] > 0 and rec["p95_ms"] >= rec["p50_ms"]
