"""Hostile-input hardening for the C++ executor server (executor/
server.cpp): the hand-written HTTP parser must bound memory, reject
malformed requests, and above all stay alive for the next request.
(The reference gets these properties for free from actix-web; a from-
scratch parser has to prove them.)"""

import socket
from pathlib import Path

import pytest

from tests.test_executor_server import RawExecutor


@pytest.fixture
def hardened(tmp_path, executor_bin):
    ex = RawExecutor(
        tmp_path, executor_bin, APP_MAX_BODY_BYTES=str(1 << 20)  # 1 MiB cap
    )
    yield ex
    ex.close()


def _raw(ex, payload: bytes, recv=True) -> bytes:
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.settimeout(10.0)
    s.connect(ex.sock)
    try:
        s.sendall(payload)
    except BrokenPipeError:
        pass  # server already rejected and closed: fine
    data = b""
    if recv:
        try:
            while True:
                chunk = s.recv(65536)
                if not chunk:
                    break
                data += chunk
        except (socket.timeout, ConnectionResetError, BrokenPipeError):
            pass
    s.close()
    return data


def _alive(ex) -> bool:
    return ex.client.get("/healthz").status_code == 200


def test_garbage_bytes_do_not_kill_server(hardened):
    _raw(hardened, b"\x00\xff\xfe not http at all\r\n\r\n")
    assert _alive(hardened)


def test_incomplete_request_then_close(hardened):
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.connect(hardened.sock)
    s.sendall(b"POST /execute HTTP/1.1\r\nContent-Le")  # hang up mid-header
    s.close()
    assert _alive(hardened)


def test_oversized_headers_rejected(hardened):
    payload = (
        b"GET /healthz HTTP/1.1\r\nX-Filler: " + b"a" * (80 * 1024) + b"\r\n\r\n"
    )
    data = _raw(hardened, payload)
    assert b"431" in data.split(b"\r\n")[0] or data == b""
    assert _alive(hardened)


def test_content_length_above_cap_rejected_without_buffering(hardened):
    # claims 1 TiB: must be refused up front, not accumulated
    head = (
        b"PUT /workspace/big.bin HTTP/1.1\r\nHost: x\r\n"
        b"Content-Length: 1099511627776\r\n\r\n"
    )
    data = _raw(hardened, head)
    assert b"413" in data.split(b"\r\n")[0]
    assert _alive(hardened)


def test_chunked_total_above_cap_rejected(hardened):
    head = (
        b"PUT /workspace/big.bin HTTP/1.1\r\nHost: x\r\n"
        b"Transfer-Encoding: chunked\r\n\r\n"
    )
    # a single declared 1 GiB chunk against the 1 MiB cap
    data = _raw(hardened, head + b"40000000\r\n")
    assert b"413" in data.split(b"\r\n")[0]
    assert _alive(hardened)


def test_malformed_chunk_size_line(hardened):
    head = (
        b"PUT /workspace/x.bin HTTP/1.1\r\nHost: x\r\n"
        b"Transfer-Encoding: chunked\r\n\r\n"
    )
    _raw(hardened, head + b"zzzz-not-hex" * 600 + b"\r\n5\r\nhello\r\n0\r\n\r\n")
    assert _alive(hardened)


def test_invalid_json_execute_is_400(hardened):
    r = hardened.client.post("/execute", content=b"{not json", headers={
        "content-type": "application/json"
    })
    assert r.status_code == 400
    assert _alive(hardened)


def test_wrong_typed_execute_body_is_400(hardened):
    r = hardened.client.post("/execute", json={"source_code": 5})
    assert r.status_code == 400
    assert _alive(hardened)


def test_negative_content_length_is_rejected(hardened):
    # strtoull("-1") wraps to 2^64-1: must hit the body cap, not allocate
    data = _raw(
        hardened,
        b"PUT /workspace/n.bin HTTP/1.1\r\nHost: x\r\nContent-Length: -1\r\n\r\n",
    )
    assert b"413" in data.split(b"\r\n")[0]
    assert _alive(hardened)


def test_rapid_connect_disconnect_storm(hardened):
    for _ in range(200):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.connect(hardened.sock)
        s.close()
    assert _alive(hardened)


def test_body_at_cap_still_accepted(hardened):
    data = b"x" * (1 << 20)  # exactly the configured cap
    r = hardened.client.put("/workspace/cap.bin", content=data)
    assert r.status_code == 204
    assert (Path(hardened.workspace) / "cap.bin").stat().st_size == len(data)


def test_invalid_utf8_output_is_lossy_not_500(hardened):
    # reference parity (String::from_utf8_lossy): raw non-UTF-8 bytes on
    # stdout/stderr must come back as U+FFFD-substituted text in valid
    # JSON, not poison the response
    r = hardened.client.post(
        "/execute",
        json={
            "source_code": (
                "import sys\n"
                "sys.stdout.buffer.write(b'\\xff\\xfeok\\x80')\n"
                "sys.stderr.buffer.write(b'\\xc3')\n"  # truncated 2-byte seq
            )
        },
    )
    assert r.status_code == 200
    body = r.json()  # decodes => the JSON is valid UTF-8
    assert body["exit_code"] == 0
    assert "ok" in body["stdout"]
    assert "\ufffd" in body["stdout"]
    assert body["stderr"].startswith("\ufffd") or "\ufffd" in body["stderr"]


def test_utf8_passthrough_unchanged(hardened):
    r = hardened.client.post(
        "/execute", json={"source_code": "print('héllo \\u4e16\\u754c')"}
    )
    assert r.status_code == 200
    assert r.json()["stdout"] == "héllo \u4e16\u754c\n"


def test_large_put_get_streams_with_bounded_rss(tmp_path, executor_bin):
    # 96 MiB round trip must not buffer the body: the server's peak RSS
    # stays far below the payload size (streamed PUT -> file, chunked GET)
    ex = RawExecutor(tmp_path, executor_bin)
    try:
        data = b"z" * (96 << 20)
        assert ex.client.put("/workspace/big.bin", content=data).status_code == 204
        got = ex.client.get("/workspace/big.bin")
        assert got.status_code == 200
        assert len(got.content) == len(data)
        hwm_kb = 0
        with open(f"/proc/{ex.proc.pid}/status") as f:
            for line in f:
                if line.startswith("VmHWM:"):
                    hwm_kb = int(line.split()[1])
        assert hwm_kb > 0 and hwm_kb < (64 << 10), f"peak RSS {hwm_kb} kB"
    finally:
        ex.close()


def test_connection_storm_5000_bounded_threads(hardened):
    # a storm far above APP_MAX_CONNECTIONS must neither kill the server
    # nor leave it wedged (acceptor gate + kernel backlog absorb it)
    for _ in range(5000):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.connect(hardened.sock)
        s.close()
    assert _alive(hardened)
