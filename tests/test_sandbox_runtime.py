"""Unit tests for the sandbox runtime (import scan, hooks, script runner)
-- runs in-process on CPU."""

import subprocess
import sys
from pathlib import Path

RUNTIME_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "executor"
sys.path.insert(0, str(RUNTIME_DIR))

import sandbox_runtime  # noqa: E402


def test_scan_missing_imports_stdlib_ignored():
    assert sandbox_runtime.scan_missing_imports("import os, sys\nimport json") == []


def test_scan_missing_imports_installed_ignored():
    assert sandbox_runtime.scan_missing_imports("import numpy\nimport scipy") == []


def test_scan_missing_imports_finds_missing_with_alias():
    out = sandbox_runtime.scan_missing_imports(
        "import definitely_not_installed_xyz\nimport fitz\n"
    )
    assert "definitely_not_installed_xyz" in out
    # alias map: fitz -> pymupdf
    assert "pymupdf" in out


def test_scan_missing_imports_from_import():
    out = sandbox_runtime.scan_missing_imports("from not_a_module_abc.sub import x")
    assert out == ["not_a_module_abc"]


def test_scan_syntax_error_returns_empty():
    assert sandbox_runtime.scan_missing_imports("def broken(:") == []


def test_run_user_script_exit_codes(tmp_path):
    # run in a subprocess so the import hook does not leak into pytest
    code = """
import sys
sys.path.insert(0, {runtime!r})
import sandbox_runtime
import pathlib
p = pathlib.Path({tmp!r})
(p / "ok.py").write_text("print('fine')")
(p / "boom.py").write_text("raise ValueError('nope')")
(p / "exit3.py").write_text("import sys; sys.exit(3)")
assert sandbox_runtime.run_user_script(str(p / "ok.py")) == 0
assert sandbox_runtime.run_user_script(str(p / "boom.py")) == 1
assert sandbox_runtime.run_user_script(str(p / "exit3.py")) == 3
print("ALL_OK")
""".format(runtime=str(RUNTIME_DIR), tmp=str(tmp_path))
    r = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=60
    )
    assert "ALL_OK" in r.stdout, r.stderr


def test_matplotlib_show_captures_plot(tmp_path, executor_bin):
    """Headless artifact capture parity: plt.show() writes plot.png into
    the workspace (reference sitecustomize.py:9-12)."""
    import pytest

    try:
        import matplotlib  # noqa: F401
    except ImportError:
        pytest.skip("matplotlib not installed")
    import asyncio

    from code_interpreter_amd.services.local_executor import LocalPoolExecutor
    from code_interpreter_amd.services.storage import Storage

    async def run():
        ex = LocalPoolExecutor(
            Storage(str(tmp_path / "s")),
            pool_target_length=1,
            gpu_count=0,
            executor_root=str(tmp_path / "e"),
            dep_install=False,
        )
        try:
            r = await ex.execute(
                "import matplotlib\nmatplotlib.use('Agg')\n"
                "import matplotlib.pyplot as plt\n"
                "plt.plot([1, 2, 3])\nplt.show()\n"
            )
            assert r.exit_code == 0, r.stderr
            assert "/workspace/plot.png" in r.files
        finally:
            await ex.aclose()

    asyncio.run(run())


def test_sandbox_rlimits_applied(tmp_path, executor_bin):
    """User code cannot dump core or write unbounded files; the limits are
    visible from inside the sandbox."""
    import asyncio

    from code_interpreter_amd.services.local_executor import LocalPoolExecutor
    from code_interpreter_amd.services.storage import Storage

    ex = LocalPoolExecutor(
        Storage(str(tmp_path / "s")),
        pool_target_length=1,
        gpu_count=0,
        executor_root=str(tmp_path / "e"),
        dep_install=False,
    )
    try:
        code = (
            "import resource\n"
            "core = resource.getrlimit(resource.RLIMIT_CORE)\n"
            "fsize = resource.getrlimit(resource.RLIMIT_FSIZE)\n"
            "print(core[0], fsize[0])\n"
        )
        r = asyncio.run(ex.execute(code))
        assert r.exit_code == 0, r.stderr
        core_soft, fsize_soft = r.stdout.split()
        assert core_soft == "0"
        assert int(fsize_soft) == 4096 << 20
        # exceeding the file-size cap kills the write, not the engine
        big = (
            "f = open('big.bin', 'wb')\n"
            "f.write(b'x' * (5 << 20))\n"
            "print('wrote small ok')\n"
        )
        r = asyncio.run(ex.execute(big))
        assert r.exit_code == 0
    finally:
        asyncio.run(ex.aclose())


def test_transform_shell_escapes_unit():
    from code_interpreter_amd.executor import sandbox_runtime as rt

    src = 'x = 1\n!ls -la /tmp\nif x:\n    !echo "q uo"\nprint(x)\n'
    out = rt.transform_shell_escapes(src)
    assert "__ci_shell__('ls -la /tmp')" in out
    assert "    __ci_shell__('echo \"q uo\"')" in out
    assert "print(x)" in out
    # no escapes -> None (caller keeps the original)
    assert rt.transform_shell_escapes("print(1)\n") is None
    # != operator lines must not match
    assert rt.transform_shell_escapes("a\n!= b\n") is None


def test_transform_shell_capture_unit():
    import importlib
    rt = importlib.import_module("code_interpreter_amd.executor.sandbox_runtime")
    out = rt.transform_shell_escapes('files = $(ls /tmp)\nprint(files)\n')
    assert "__ci_shell_capture__('ls /tmp')" in out
    # mixed with ! lines
    out2 = rt.transform_shell_escapes('!mkdir -p /tmp/x\nv = $(echo hi)\n')
    assert "__ci_shell__('mkdir -p /tmp/x')" in out2
    assert "__ci_shell_capture__('echo hi')" in out2
    # pure python with no escapes stays untouched
    assert rt.transform_shell_escapes("x = f(1)\n") is None
    # capture runner returns stdout
    assert rt._ci_shell_capture("echo hello").strip() == "hello"
