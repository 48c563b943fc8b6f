"""Every shipped example payload runs through the real service HTTP API
(the reference ships examples/ as sample payloads for its e2e client;
here the runner examples/run.py is part of the repo, so the payloads are
kept executable by CI)."""

from pathlib import Path

import pytest

EXAMPLES = Path(__file__).resolve().parent.parent / "examples"


def _execute(http_client, name, files=None):
    r = http_client.post(
        "/v1/execute",
        json={
            "source_code": (EXAMPLES / name).read_text(),
            "files": files or {},
        },
    )
    assert r.status_code == 200, r.text
    return r.json()


def test_fib(http_client):
    out = _execute(http_client, "fib.py")
    assert out["exit_code"] == 0
    assert out["stdout"].splitlines()[-1] == "29 514229"


def test_escaping_roundtrip(http_client):
    out = _execute(http_client, "escaping.py")
    assert out["exit_code"] == 0
    assert "single 'quotes' inside double" in out["stdout"]
    assert 'double "quotes" inside single' in out["stdout"]
    assert "newline:\nnext line" in out["stdout"]
    assert "backslash: \\ and raw $DOLLAR ${BRACES} `backticks`" in out["stdout"]


def test_crash_returns_traceback(http_client):
    out = _execute(http_client, "crash.py")
    assert out["exit_code"] != 0
    assert "IndexError" in out["stderr"]


def test_ls_fresh_workspace(http_client):
    out = _execute(http_client, "ls.py")
    assert out["exit_code"] == 0
    assert out["stdout"].splitlines()[1] == "[]"  # empty per-execution cwd


def test_tcp_loopback(http_client):
    out = _execute(http_client, "tcp.py")
    assert out["exit_code"] == 0
    assert "echoed: ping over loopback" in out["stdout"]


def test_files_scan(http_client):
    out = _execute(http_client, "files.py")
    assert out["exit_code"] == 0
    assert out["stdout"] == "written inside the sandbox\n"
    assert any(p.endswith("note.txt") for p in out["files"])


def test_write_then_read_roundtrip(http_client):
    first = _execute(http_client, "hello_world_write_file.py")
    assert first["exit_code"] == 0
    (path, digest), = [
        (p, h) for p, h in first["files"].items() if p.endswith("example.txt")
    ]
    second = _execute(http_client, "hello_world_read_file.py", files={path: digest})
    assert second["exit_code"] == 0
    assert second["stdout"] == "hello from a previous execution\n"


def test_using_imports(http_client):
    out = _execute(http_client, "using_imports.py")
    assert out["exit_code"] == 0, out["stderr"]
    assert "p =" in out["stdout"]


@pytest.mark.slow
def test_mixed_workload_under_load(http_client):
    """BASELINE config 5: using_imports-style + files-style payloads
    concurrently -- the import scan, workspace staging and changed-file
    scan hold up under load with zero errors."""
    import concurrent.futures

    files_payload = (EXAMPLES / "files.py").read_text()
    imports_payload = (
        "import numpy as np\n"
        "import pandas as pd\n"
        "from scipy import stats\n"
        "x = np.arange(100, dtype=float)\n"
        "print(float(pd.Series(x).mean()), float(stats.sem(x)))\n"
    )

    def one(i):
        payload = files_payload if i % 2 == 0 else imports_payload
        r = http_client.post("/v1/execute", json={"source_code": payload})
        assert r.status_code == 200, r.text
        out = r.json()
        assert out["exit_code"] == 0, out["stderr"]
        if i % 2 == 0:
            assert any(p.endswith("note.txt") for p in out["files"])
        return i

    with concurrent.futures.ThreadPoolExecutor(max_workers=8) as pool:
        done = list(pool.map(one, range(40)))
    assert len(done) == 40


def test_analysis_example_cpu_fallback(http_client):
    # the r02 showcase chain must give identical values on plain CPU
    # numpy (no GPU in this env): the routing layer's fallback contract
    out = _execute(http_client, "analysis-gpu.py")
    assert out["exit_code"] == 0, out["stderr"]
    assert "outliers:" in out["stdout"]
    assert "kind: ndarray" in out["stdout"]


def test_analytics_surface(http_client):
    out = _execute(http_client, "analytics.py")
    assert out["exit_code"] == 0, out["stderr"]
    lines = out["stdout"].splitlines()
    assert lines[2].startswith("hist total: 2000000")
    assert lines[3].endswith("True")
    assert lines[4].endswith("True")
    assert lines[5].endswith("True")
    assert lines[6].endswith("True")
    assert lines[7].endswith("True")
