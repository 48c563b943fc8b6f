"""RemoteBackend <-> daemon wire agreement for EVERY op: the sender's
header keys must match what hipd's dispatch reads. Runs the real
RemoteBackend against the real Connection dispatch (unix socket, fake
in-process _hipops) -- catches key-name drift that otherwise only a GPU
box would reveal."""

import socket
import sys
import threading
from pathlib import Path

import numpy as np
import pytest

OPS_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"

from test_hipd_ownership import FakeHipops  # noqa: E402


@pytest.fixture
def backend(monkeypatch, tmp_path):
    fake = FakeHipops()
    monkeypatch.setitem(sys.modules, "_hipops", fake)
    monkeypatch.syspath_prepend(str(OPS_DIR))
    sys.modules.pop("hipd", None)
    import hipd

    monkeypatch.setattr(hipd, "_hipops", fake)
    path = str(tmp_path / "gpu.sock")
    server = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    server.bind(path)
    server.listen(4)
    conns = []

    def acceptor():
        while True:
            try:
                c, _ = server.accept()
            except OSError:
                return
            conn = hipd.Connection(c)
            conn.start()
            conns.append(conn)

    t = threading.Thread(target=acceptor, daemon=True)
    t.start()
    import hipnp

    b = hipnp.RemoteBackend(path)
    yield b, fake
    try:
        b._sock.close()
    except OSError:
        pass
    server.close()
    sys.modules.pop("hipd", None)


def test_every_op_round_trips(backend):
    b, fake = backend
    a = np.arange(32, dtype=np.float64)

    h = b.upload(a)
    out = np.empty_like(a)
    b.download(h, out)
    np.testing.assert_array_equal(out, a)

    h2 = b.unary(h, 0, 1, 32)
    h3 = b.binary(h, h2, 0, 1, 32)
    h4 = b.binary_scalar(h, 2.0, 2, 1, 32)
    assert b.sum(h, 1, 32, 0) == 0.0  # FakeHipops stub value
    hg = b.gemm(h, h2, 4, 4, 4, 1)
    hr = b.reduce_axis(h, 1, 4, 4, 2, 0)
    hb2 = b.gemm_batched(h, h2, 2, 2, 2, 2, 1)
    assert b.argminmax(h, 1, 32, 1) == 0
    hs = b.sort(h, 1, 32, 0)
    hsv, hsi = b.sort(h, 1, 32, 1)
    h2d = b.sort2d(h, 1, 4, 8, 0)
    h2v, h2i = b.sort2d(h, 1, 4, 8, 1)
    ht = b.transpose(h, 1, 4, 8)
    hc = b.cumsum(h, 1, 32)
    hc2 = b.cumsum2d(h, 1, 4, 8)
    hd = b.diff(h, 1, 1, 32)
    hss = b.searchsorted(h, 32, h2, 32, 1, 0)
    raw = b.download_slice(h, 8, 16)
    assert len(raw) == 16
    raw2 = b.download_strided(h, 0, 16, 8, 4)
    assert len(raw2) == 32
    hist = b.histogram(h, 1, 32, 0.0, 31.0, 8, 1)
    assert len(hist) == (8 + 3) * 8
    dst = b.alloc(256)
    b.copy_d2d(dst, 0, h, 0, 256)
    for hh in (h, h2, h3, h4, hg, hr, hb2, hs, hsv, hsi, h2d, h2v, h2i,
               ht, hc, hc2, hd, hss, dst):
        b.free(hh)
    # all buffers released through the daemon (cleanup parity)
    assert b.mem_info() is not None
