"""Direct unit tests for the kubectl CLI wrapper using a stub binary (no
cluster): argv construction, JSON-output handling, stdin manifests, and
the RuntimeError-on-failure contract the retry layer depends on."""

import asyncio
import json
import os
import stat
import sys

import pytest

from code_interpreter_amd.services.kubectl import Kubectl


@pytest.fixture
def stub(tmp_path):
    """A fake kubectl that records its argv + stdin and prints a canned
    response controlled by env vars."""
    record = tmp_path / "record.json"
    binpath = tmp_path / "kubectl-stub"
    binpath.write_text(
        "#!%s\n" % sys.executable
        + "import json, os, sys\n"
        + "data = sys.stdin.read() if not sys.stdin.isatty() else ''\n"
        + "json.dump({'argv': sys.argv[1:], 'stdin': data},"
        + " open(%r, 'w'))\n" % str(record)
        + "sys.stdout.write(os.environ.get('STUB_OUT', '{}'))\n"
        + "sys.exit(int(os.environ.get('STUB_RC', '0')))\n"
    )
    binpath.chmod(binpath.stat().st_mode | stat.S_IEXEC)
    return Kubectl(kubectl_bin=str(binpath)), record


def test_kwargs_become_flags(stub, monkeypatch):
    k, record = stub
    monkeypatch.setenv("STUB_OUT", '{"kind": "Pod"}')
    out = asyncio.run(k.get("pod", "mypod", output_watch_events=True, timeout="60s"))
    assert out == {"kind": "Pod"}
    rec = json.loads(record.read_text())
    assert rec["argv"][:3] == ["get", "pod", "mypod"]
    assert "--output-watch-events=true" in rec["argv"]
    assert "--timeout=60s" in rec["argv"]
    assert rec["argv"][-1] == "--output=json"  # JSON subcommand


def test_non_json_subcommand_returns_text(stub, monkeypatch):
    k, record = stub
    monkeypatch.setenv("STUB_OUT", "pod/mypod condition met")
    out = asyncio.run(k.wait("pod/mypod", **{"for": "condition=Ready"}))
    assert out == "pod/mypod condition met"
    assert "--output=json" not in json.loads(record.read_text())["argv"]


def test_body_piped_via_stdin(stub, monkeypatch):
    k, record = stub
    monkeypatch.setenv("STUB_OUT", '{"metadata": {"name": "x"}}')
    manifest = {"apiVersion": "v1", "kind": "Pod"}
    asyncio.run(k.create(body=manifest))
    rec = json.loads(record.read_text())
    assert rec["argv"][:1] == ["create"]
    assert "-f" in rec["argv"] and "-" in rec["argv"]
    assert json.loads(rec["stdin"]) == manifest


def test_nonzero_exit_raises_runtimeerror(stub, monkeypatch):
    k, _ = stub
    monkeypatch.setenv("STUB_RC", "1")
    with pytest.raises(RuntimeError, match="kubectl delete failed"):
        asyncio.run(k.delete("pod", "gone"))


def test_none_kwargs_skipped(stub, monkeypatch):
    k, record = stub
    monkeypatch.setenv("STUB_OUT", "{}")
    asyncio.run(k.get("pods", namespace=None))
    assert not any(a.startswith("--namespace") for a in json.loads(record.read_text())["argv"])
