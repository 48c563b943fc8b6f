"""Per-engine GPU compute daemon.

Holds the engine's ONE HIP context (per-process context creation
serializes in the kernel driver at ~13/s -- measured, see
profiles/NOTES.md), a shared device-memory free-list, and the pinned
staging buffers. Sandbox children don't own a GPU context: their numpy
ops RPC here over a unix socket (hipnp's remote backend), so a fresh
sandbox costs a fork, and repeated same-shape allocations across requests
hit the warm pool.

One thread per connection; _hipops releases the GIL around device work.
All handles opened by a connection are freed when it closes (sandbox
exit == resource cleanup).

Protocol (little-endian): [u32 header_len][json header][payload bytes]
both directions; header carries "plen" when a payload follows.

Usage: python hipd.py --socket PATH   (exits 3 if no GPU is visible)
"""

import argparse
import json
import os
import socket
import struct
import sys
import threading

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
if OPS_DIR not in sys.path:
    sys.path.insert(0, OPS_DIR)

import _hipops


def read_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray(n)
    view = memoryview(buf)
    got = 0
    while got < n:
        r = sock.recv_into(view[got:], n - got)
        if r == 0:
            raise ConnectionError("peer closed")
        got += r
    return bytes(buf)


def recv_msg(sock):
    (hlen,) = struct.unpack("<I", read_exact(sock, 4))
    header = json.loads(read_exact(sock, hlen))
    payload = b""
    plen = header.get("plen", 0)
    if plen:
        payload = read_exact(sock, plen)
    return header, payload


def send_msg(sock, header: dict, payload: bytes = b"") -> None:
    if payload:
        header = {**header, "plen": len(payload)}
    hb = json.dumps(header).encode()
    sock.sendall(struct.pack("<I", len(hb)) + hb + payload)


class Connection(threading.Thread):
    def __init__(self, sock):
        super().__init__(daemon=True)
        self.sock = sock
        self.handles = set()

    def _own(self, h):
        """Ownership check on every op that consumes a handle: handles are
        small sequential integers, so without this a sandbox could
        enumerate other connections' handles and read/corrupt buffers
        belonging to concurrent executions (the reference's isolation
        unit is the pod; ours is the connection)."""
        if h not in self.handles:
            raise ValueError(f"handle {h} not owned by this connection")
        return h

    def run(self):
        try:
            while True:
                header, payload = recv_msg(self.sock)
                try:
                    resp, out_payload = self.dispatch(header, payload)
                except Exception as e:  # per-op errors back to client
                    resp, out_payload = {"ok": False, "error": str(e)}, b""
                send_msg(self.sock, resp, out_payload)
        except (ConnectionError, OSError):
            pass
        finally:
            for h in self.handles:
                try:
                    _hipops.free(h)
                except Exception:
                    pass
            try:
                self.sock.close()
            except OSError:
                pass

    def dispatch(self, m: dict, payload: bytes):
        op = m["op"]
        if op == "ping":
            return {"ok": True}, b""
        if op == "upload":
            h = _hipops.upload(payload)
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "download":
            out = bytearray(m["nbytes"])
            _hipops.download(self._own(m["h"]), out)
            return {"ok": True}, bytes(out)
        if op == "upload_shm":
            # large-transfer path: payload arrives as a /dev/shm file the
            # client wrote (one memcpy each side instead of a socket
            # stream); path is restricted to /dev/shm
            path = m["path"]
            if not path.startswith("/dev/shm/"):
                raise ValueError("upload_shm path must be under /dev/shm")
            import mmap as _mmap

            with open(path, "rb") as f:
                mapped = _mmap.mmap(f.fileno(), m["nbytes"],
                                    prot=_mmap.PROT_READ)
                try:
                    h = _hipops.upload(memoryview(mapped))
                finally:
                    mapped.close()
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "download_shm":
            path = m["path"]
            if not path.startswith("/dev/shm/"):
                raise ValueError("download_shm path must be under /dev/shm")
            import mmap as _mmap

            with open(path, "r+b") as f:
                f.truncate(m["nbytes"])
                mapped = _mmap.mmap(f.fileno(), m["nbytes"])
                try:
                    _hipops.download(self._own(m["h"]), memoryview(mapped))
                    mapped.flush()
                finally:
                    mapped.close()
            return {"ok": True}, b""
        if op == "alloc":
            h = _hipops.alloc(m["nbytes"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "free":
            self._own(m["h"])
            self.handles.discard(m["h"])
            _hipops.free(m["h"])
            return {"ok": True}, b""
        if op == "rand":
            h = _hipops.rand(m["n"], m["dtype"], m["seed"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "randn":
            h = _hipops.randn(m["n"], m["seed"], m["mu"], m["sigma"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "convert":
            h = _hipops.convert(self._own(m["h"]), m["src"], m["dst"], m["n"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "unary":
            h = _hipops.unary(self._own(m["h"]), m["uop"], m["dtype"], m["n"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "binary":
            h = _hipops.binary(self._own(m["ha"]), self._own(m["hb"]), m["bop"], m["dtype"], m["n"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "binary_scalar":
            h = _hipops.binary_scalar(
                self._own(m["h"]), m["scalar"], m["bop"], m["dtype"], m["n"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "sum":
            v = _hipops.sum(self._own(m["h"]), m["dtype"], m["n"], m["square"])
            return {"ok": True, "value": v}, b""
        if op == "cumsum":
            h = _hipops.cumsum(self._own(m["h"]), m["dtype"], m["n"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "download_strided":
            data = _hipops.download_strided(
                self._own(m["h"]), m["off"], m["stride"], m["esz"],
                m["count"]
            )
            return {"ok": True}, data
        if op == "download_slice":
            data = _hipops.download_slice(
                self._own(m["h"]), m["off"], m["nbytes"]
            )
            return {"ok": True}, data
        if op == "searchsorted":
            h = _hipops.searchsorted(
                self._own(m["ha"]), m["n"], self._own(m["hv"]), m["m"],
                m["dtype"], m["right"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "diff":
            h = _hipops.diff(
                self._own(m["h"]), m["dtype"], m["outer"], m["inner"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "cumsum2d":
            h = _hipops.cumsum2d(
                self._own(m["h"]), m["dtype"], m["rows"], m["cols"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "copy_d2d":
            _hipops.copy_d2d(
                self._own(m["hd"]), m["doff"], self._own(m["hs"]),
                m["soff"], m["nbytes"]
            )
            return {"ok": True}, b""
        if op == "transpose":
            h = _hipops.transpose(
                self._own(m["h"]), m["dtype"], m["rows"], m["cols"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "sort2d":
            r = _hipops.sort2d(
                self._own(m["h"]), m["dtype"], m["rows"], m["cols"],
                m["want_idx"]
            )
            if m["want_idx"]:
                h, hi = r
                self.handles.add(h)
                self.handles.add(hi)
                return {"ok": True, "h": h, "hi": hi}, b""
            self.handles.add(r)
            return {"ok": True, "h": r}, b""
        if op == "sort":
            r = _hipops.sort(
                self._own(m["h"]), m["dtype"], m["n"], m["want_idx"]
            )
            if m["want_idx"]:
                h, hi = r
                self.handles.add(h)
                self.handles.add(hi)
                return {"ok": True, "h": h, "hi": hi}, b""
            self.handles.add(r)
            return {"ok": True, "h": r}, b""
        if op == "mask_logic":
            hb = m.get("hb", 0)
            if hb:
                self._own(hb)
            h = _hipops.mask_logic(self._own(m["ha"]), hb, m["n"], m["lop"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "histogram":
            data = _hipops.histogram(
                self._own(m["h"]), m["dtype"], m["n"], m["lo"], m["hi"],
                m["bins"], m.get("exact", 0)
            )
            return {"ok": True}, data
        if op == "extract_range":
            count, data = _hipops.extract_range(
                self._own(m["h"]), m["dtype"], m["n"], m["lo"], m["hi"],
                m["cap"]
            )
            return {"ok": True, "count": count}, data
        if op == "compare":
            hb = m.get("hb", 0)
            if hb:
                self._own(hb)
            h = _hipops.compare(
                self._own(m["h"]), m["dtype"], m["n"], m["cmp"], hb,
                m.get("scalar", 0.0)
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "where":
            ha, hb = m.get("ha", 0), m.get("hb", 0)
            if ha:
                self._own(ha)
            if hb:
                self._own(hb)
            h = _hipops.where(
                self._own(m["hm"]), m["dtype"], m["n"], ha,
                m.get("sa", 0.0), hb, m.get("sb", 0.0)
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "masked_fill":
            _hipops.masked_fill(
                self._own(m["h"]), self._own(m["hm"]), m["dtype"], m["n"],
                m["value"]
            )
            return {"ok": True}, b""
        if op == "mask_count":
            v = _hipops.mask_count(self._own(m["hm"]), m["n"])
            return {"ok": True, "value": v}, b""
        if op == "binary_bcast":
            h = _hipops.binary_bcast(
                self._own(m["ha"]), self._own(m["hb"]), m["bop"], m["dtype"],
                m["outer"], m["inner"], m["mode"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "argminmax":
            v = _hipops.argminmax(
                self._own(m["h"]), m["dtype"], m["n"], m["maxop"]
            )
            return {"ok": True, "value": v}, b""
        if op == "reduce_axis":
            h = _hipops.reduce_axis(
                self._own(m["h"]), m["dtype"], m["outer"], m["red"],
                m["inner"], m["mode"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "gemm_batched":
            h = _hipops.gemm_batched(
                self._own(m["ha"]), self._own(m["hb"]), m["batch"], m["m"],
                m["n"], m["k"], m["dtype"]
            )
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "gemm":
            h = _hipops.gemm(self._own(m["ha"]), self._own(m["hb"]), m["m"], m["n"], m["k"], m["dtype"])
            self.handles.add(h)
            return {"ok": True, "h": h}, b""
        if op == "sync":
            _hipops.synchronize()
            return {"ok": True}, b""
        if op == "mem_info":
            info = _hipops.mem_info()
            return {"ok": True, "info": list(info)}, b""
        raise ValueError(f"unknown op {op!r}")


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--socket", required=True)
    parser.add_argument("--device", type=int, default=0)
    args = parser.parse_args()

    # clean exit on SIGTERM (engine shutdown broadcast): run atexit and
    # profiler finalizers -- a SIGKILLed daemon loses any rocprofv3
    # output it was asked to collect
    import signal as _signal

    _signal.signal(_signal.SIGTERM, lambda *_: sys.exit(0))

    if not _hipops.is_available():
        sys.exit(3)
    _hipops.init(args.device)
    # warm the hot kernels + staging once, before accepting connections
    h = _hipops.rand(1 << 20, 1, 7)
    _hipops.sum(h, 1, 1 << 20, 1)
    _hipops.free(h)
    _hipops.synchronize()

    try:
        os.unlink(args.socket)
    except FileNotFoundError:
        pass
    server = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    server.bind(args.socket)
    server.listen(128)
    # readiness marker for the spawning server
    print("READY", flush=True)
    while True:
        conn, _ = server.accept()
        Connection(conn).start()


if __name__ == "__main__":
    main()
