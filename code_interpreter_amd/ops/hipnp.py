"""numpy -> gfx950 HIP kernel routing for sandboxed user code.

Two mechanisms, installed by the sandbox runtime (executor/sandbox_runtime
.py) when a GPU is visible:

1. patched module-level entry points (numpy.random.{rand, random,
   random_sample, uniform, randn, standard_normal, normal}, numpy.matmul,
   numpy.dot, numpy.square, numpy.sum, numpy.{sqrt,exp,log,sin,cos,tanh,
   abs}): above a size threshold the work runs on the MI355X through
   _hipops and the result stays device-resident;
2. DeviceArray: a duck array (NEP 13/18 __array_ufunc__ +
   __array_function__) so follow-on numpy calls on a device-resident
   result keep running on the GPU: 26 elementwise unaries, binary ops
   (incl. row/column broadcasting), full and axis-wise reductions (plus
   the nan* family), argmax/argmin, clip, matmul (2-D / matvec /
   equal-batch 3-D / 1-D dot / pad-to-256 bf16), sort/argsort/
   partition/unique/searchsorted (stable radix sort, 2-D both axes),
   median/quantile/percentile (flat, per-row/column, array q),
   histogram, cumsum (flat/2-D), transpose/.T, reshape/ravel, diff,
   cov/corrcoef, einsum common contractions, outer/trace, linalg.norm,
   concatenate/stack family, nan_to_num, isclose/allclose, ptp,
   weighted average, device boolean masks (comparisons, np.where,
   x[mask] = v, popcount), in-place mutation and ufunc out=. Any
   unsupported operation transparently materializes to a host ndarray
   and computes numpy's own result -- identical values either way.

Everything computes the same values user code would get on the CPU (same
dtype; RNG is Philox instead of MT19937 -- a documented backend change,
seeded from numpy's global RNG so np.random.seed still fixes the stream).

Compute backends: inside an engine, sandbox children talk to the per-GPU
compute daemon (ops/hipd.py, socket from APP_GPU_SERVICE) -- one HIP
context per GPU, shared memory pool, fork-cheap sandboxes (per-process
HIP context creation serializes in the driver at ~13/s). Without a
daemon, the local backend owns the context directly through _hipops.

This file is imported standalone (sys.path) inside sandbox children; it
must not import the control-plane package or torch.
"""

import os
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
if _OPS_DIR not in sys.path:
    sys.path.insert(0, _OPS_DIR)

import numpy as _np

import json as _json
import socket as _socket
import struct as _struct
import threading as _threading


def _load_hipops():
    """Import _hipops ON DEMAND, never at module import.

    Sandbox children that talk to the GPU daemon (the serving hot path)
    need no local HIP at all; importing _hipops would map the ~0.3-CPU-s
    HIP userspace stack per forked child (the r01 CFS-throttle cliff)
    AND bind our kernels to whichever libamdhip64.so.7 instance is
    already loaded -- if that instance was loaded before the fork, kernel
    launches through it segfault (isolated by scripts/torch_case.py:
    torch, which always uses its own bundled runtime loaded post-fork,
    is unaffected; our kernels through a pre-fork-loaded runtime die).
    Loading lazily in exactly the process that launches kernels keeps
    every binding post-fork and the hot path mapping-free."""
    import _hipops as m

    return m


class LocalBackend:
    """Direct _hipops calls: this process owns the HIP context."""

    name = "local"

    def __init__(self):
        self._m = _load_hipops()
        self._m.init(0)

    def __getattr__(self, item):
        return getattr(self._m, item)


class GpuBackendLost(RuntimeError):
    """The GPU daemon connection died while device state was live; the
    sandbox cannot transparently recover (its handles are gone). The
    runner maps this to exit code 113, which the control plane treats as
    an infrastructure failure and retries the whole execution in a fresh
    sandbox (the same semantics as the reference's pod-death retry)."""


class RemoteBackend:
    """RPC to the engine's GPU daemon (ops/hipd.py) over a unix socket.
    On connection loss with no live device handles (e.g. a pre-warmed
    child whose daemon restarted) it transparently reconnects and
    retries; with live handles it raises GpuBackendLost.

    One connection (and one lock) per sandbox process, BY DESIGN: the
    daemon's handle-ownership boundary is the connection, so a
    per-thread connection pool would make device arrays unusable across
    the user's threads (each thread's handles would be invisible to the
    others). The cost is that a sandbox's device ops serialize at the
    RPC layer even if user code is multi-threaded -- an accepted trade
    for cross-request isolation; compute itself is already serialized
    per GPU by the daemon's single HIP context."""

    name = "remote"

    def __init__(self, path: str):
        self._path = path
        self._lock = _threading.Lock()
        self._live_handles = 0
        self._connect()
        self._call({"op": "ping"})

    def _connect(self):
        self._sock = _socket.socket(_socket.AF_UNIX, _socket.SOCK_STREAM)
        self._sock.connect(self._path)

    def _send(self, header: dict, payload=None) -> None:
        if payload is not None:
            header = {**header, "plen": len(payload)}
        hb = _json.dumps(header).encode()
        self._sock.sendall(_struct.pack("<I", len(hb)) + hb)
        if payload is not None and len(payload):
            self._sock.sendall(payload)

    def _read_exact_into(self, view) -> None:
        got = 0
        n = len(view)
        while got < n:
            r = self._sock.recv_into(view[got:], n - got)
            if r == 0:
                raise ConnectionError("gpu daemon closed the connection")
            got += r

    def _recv(self, out_buffer=None) -> dict:
        head = bytearray(4)
        self._read_exact_into(memoryview(head))
        (hlen,) = _struct.unpack("<I", bytes(head))
        hb = bytearray(hlen)
        self._read_exact_into(memoryview(hb))
        header = _json.loads(bytes(hb))
        plen = header.get("plen", 0)
        if plen:
            if out_buffer is not None and len(out_buffer) == plen:
                self._read_exact_into(memoryview(out_buffer).cast("B"))
            else:
                tmp = bytearray(plen)
                self._read_exact_into(memoryview(tmp))
                header["_payload"] = bytes(tmp)
        if not header.get("ok", False):
            raise RuntimeError(header.get("error", "gpu daemon error"))
        return header

    def _reconnect_and_retry(self, header, payload, out_buffer):
        if self._live_handles > 0:
            raise GpuBackendLost(
                "gpu daemon connection lost with live device handles"
            )
        import time as _time

        wait = float(os.environ.get("APP_GPU_SERVICE_WAIT", "10"))
        deadline = _time.monotonic() + wait
        while True:
            try:
                self._connect()
                self._send(header, payload)
                return self._recv(out_buffer)
            except (ConnectionError, OSError):
                if _time.monotonic() >= deadline:
                    raise GpuBackendLost("gpu daemon unreachable")
                _time.sleep(0.1)

    def _call(self, header: dict, payload=None, out_buffer=None) -> dict:
        import time as _time

        t0 = _time.perf_counter()
        with self._lock:
            try:
                self._send(header, payload)
                out = self._recv(out_buffer)
            except (ConnectionError, OSError):
                out = self._reconnect_and_retry(header, payload, out_buffer)
        if "h" in out:
            self._live_handles += 1
        elif header.get("op") == "free":
            self._live_handles = max(0, self._live_handles - 1)
        dt = (_time.perf_counter() - t0) * 1000
        RPC_STATS["ms"] += dt
        RPC_STATS["n"] += 1
        per_op = RPC_STATS.setdefault("per_op", {})
        per_op[header["op"]] = round(per_op.get(header["op"], 0.0) + dt, 2)
        return out

    # _hipops-compatible surface ----------------------------------------
    def is_available(self) -> bool:
        return True

    # Large transfers go through /dev/shm instead of the socket: the
    # stream path double-copies at unix-socket speed (~2 GB/s), while a
    # shm file costs one memcpy on each side and the daemon's pinned
    # staging runs at PCIe rate. Threshold: small messages are cheaper
    # inline.
    SHM_MIN_BYTES = 8 << 20
    SHM_DIR = "/dev/shm"

    def _shm_file(self, nbytes: int):
        import tempfile

        fd, path = tempfile.mkstemp(prefix="hipnp-", dir=self.SHM_DIR)
        try:
            os.ftruncate(fd, nbytes)
        except OSError:
            os.close(fd)
            os.unlink(path)
            raise
        return fd, path

    def upload(self, buffer):
        mv = memoryview(buffer).cast("B")
        if mv.nbytes >= self.SHM_MIN_BYTES and os.path.isdir(self.SHM_DIR):
            try:
                fd, path = self._shm_file(mv.nbytes)
            except OSError:
                return self._call({"op": "upload"}, payload=mv)["h"]
            try:
                with os.fdopen(fd, "wb") as f:
                    f.write(mv)
                return self._call(
                    {"op": "upload_shm", "path": path, "nbytes": mv.nbytes}
                )["h"]
            finally:
                try:
                    os.unlink(path)  # daemon already consumed it
                except FileNotFoundError:
                    pass
        return self._call({"op": "upload"}, payload=mv)["h"]

    def download(self, h, out) -> None:
        mv = memoryview(out).cast("B")
        if mv.nbytes >= self.SHM_MIN_BYTES and os.path.isdir(self.SHM_DIR):
            try:
                fd, path = self._shm_file(mv.nbytes)
            except OSError:
                self._call(
                    {"op": "download", "h": h, "nbytes": mv.nbytes},
                    out_buffer=mv,
                )
                return
            try:
                os.close(fd)
                self._call(
                    {"op": "download_shm", "h": h, "path": path,
                     "nbytes": mv.nbytes}
                )
                with open(path, "rb") as f:
                    f.readinto(mv)
                return
            finally:
                try:
                    os.unlink(path)
                except FileNotFoundError:
                    pass
        self._call(
            {"op": "download", "h": h, "nbytes": mv.nbytes},
            out_buffer=mv,
        )

    def alloc(self, nbytes):
        return self._call({"op": "alloc", "nbytes": nbytes})["h"]

    def free(self, h) -> None:
        self._call({"op": "free", "h": h})

    def rand(self, n, dtype, seed):
        return self._call({"op": "rand", "n": n, "dtype": dtype, "seed": seed})["h"]

    def randn(self, n, seed, mu, sigma):
        return self._call(
            {"op": "randn", "n": n, "seed": seed, "mu": mu, "sigma": sigma}
        )["h"]

    def convert(self, h, src, dst, n):
        return self._call({"op": "convert", "h": h, "src": src, "dst": dst, "n": n})["h"]

    def unary(self, h, uop, dtype, n):
        return self._call(
            {"op": "unary", "h": h, "uop": uop, "dtype": dtype, "n": n}
        )["h"]

    def binary(self, ha, hb, bop, dtype, n):
        return self._call(
            {"op": "binary", "ha": ha, "hb": hb, "bop": bop, "dtype": dtype, "n": n}
        )["h"]

    def binary_scalar(self, h, scalar, bop, dtype, n):
        return self._call(
            {
                "op": "binary_scalar",
                "h": h,
                "scalar": scalar,
                "bop": bop,
                "dtype": dtype,
                "n": n,
            }
        )["h"]

    def sum(self, h, dtype, n, square):
        return self._call(
            {"op": "sum", "h": h, "dtype": dtype, "n": n, "square": square}
        )["value"]

    def gemm(self, ha, hb, m, n, k, dtype):
        return self._call(
            {"op": "gemm", "ha": ha, "hb": hb, "m": m, "n": n, "k": k, "dtype": dtype}
        )["h"]

    def cumsum(self, h, dtype, n):
        return self._call(
            {"op": "cumsum", "h": h, "dtype": dtype, "n": n}
        )["h"]

    def download_slice(self, h, off, nbytes):
        out = self._call(
            {"op": "download_slice", "h": h, "off": off, "nbytes": nbytes}
        )
        return out["_payload"]

    def download_strided(self, h, off, stride, esz, count):
        out = self._call(
            {"op": "download_strided", "h": h, "off": off, "stride": stride,
             "esz": esz, "count": count}
        )
        return out["_payload"]

    def sort(self, h, dtype, n, want_idx):
        out = self._call(
            {"op": "sort", "h": h, "dtype": dtype, "n": n,
             "want_idx": want_idx}
        )
        if want_idx:
            return out["h"], out["hi"]
        return out["h"]

    def searchsorted(self, ha, n, hv, m, dtype, right):
        return self._call(
            {"op": "searchsorted", "ha": ha, "n": n, "hv": hv, "m": m,
             "dtype": dtype, "right": right}
        )["h"]

    def diff(self, h, dtype, outer, inner):
        return self._call(
            {"op": "diff", "h": h, "dtype": dtype, "outer": outer,
             "inner": inner}
        )["h"]

    def cumsum2d(self, h, dtype, rows, cols):
        return self._call(
            {"op": "cumsum2d", "h": h, "dtype": dtype, "rows": rows,
             "cols": cols}
        )["h"]

    def copy_d2d(self, hd, doff, hs, soff, nbytes):
        self._call(
            {"op": "copy_d2d", "hd": hd, "doff": doff, "hs": hs,
             "soff": soff, "nbytes": nbytes}
        )

    def transpose(self, h, dtype, rows, cols):
        return self._call(
            {"op": "transpose", "h": h, "dtype": dtype, "rows": rows,
             "cols": cols}
        )["h"]

    def sort2d(self, h, dtype, rows, cols, want_idx):
        out = self._call(
            {"op": "sort2d", "h": h, "dtype": dtype, "rows": rows,
             "cols": cols, "want_idx": want_idx}
        )
        if want_idx:
            return out["h"], out["hi"]
        return out["h"]

    def mask_logic(self, ha, hb, n, lop):
        return self._call(
            {"op": "mask_logic", "ha": ha, "hb": hb, "n": n, "lop": lop}
        )["h"]

    def histogram(self, h, dtype, n, lo, hi, bins, exact=0):
        out = self._call(
            {"op": "histogram", "h": h, "dtype": dtype, "n": n, "lo": lo,
             "hi": hi, "bins": bins, "exact": exact}
        )
        return out["_payload"]

    def extract_range(self, h, dtype, n, lo, hi, cap):
        out = self._call(
            {"op": "extract_range", "h": h, "dtype": dtype, "n": n,
             "lo": lo, "hi": hi, "cap": cap}
        )
        return out["count"], out.get("_payload", b"")

    def compare(self, h, dtype, n, cmp, hb, scalar):
        return self._call(
            {"op": "compare", "h": h, "dtype": dtype, "n": n, "cmp": cmp,
             "hb": hb, "scalar": scalar}
        )["h"]

    def where(self, hm, dtype, n, ha, sa, hb, sb):
        return self._call(
            {"op": "where", "hm": hm, "dtype": dtype, "n": n, "ha": ha,
             "sa": sa, "hb": hb, "sb": sb}
        )["h"]

    def masked_fill(self, h, hm, dtype, n, value):
        self._call(
            {"op": "masked_fill", "h": h, "hm": hm, "dtype": dtype, "n": n,
             "value": value}
        )

    def mask_count(self, hm, n):
        return self._call({"op": "mask_count", "hm": hm, "n": n})["value"]

    def binary_bcast(self, ha, hb, bop, dtype, outer, inner, mode):
        return self._call(
            {"op": "binary_bcast", "ha": ha, "hb": hb, "bop": bop,
             "dtype": dtype, "outer": outer, "inner": inner, "mode": mode}
        )["h"]

    def argminmax(self, h, dtype, n, maxop):
        return self._call(
            {"op": "argminmax", "h": h, "dtype": dtype, "n": n,
             "maxop": maxop}
        )["value"]

    def reduce_axis(self, h, dtype, outer, red, inner, mode):
        return self._call(
            {"op": "reduce_axis", "h": h, "dtype": dtype, "outer": outer,
             "red": red, "inner": inner, "mode": mode}
        )["h"]

    def gemm_batched(self, ha, hb, batch, m, n, k, dtype):
        return self._call(
            {"op": "gemm_batched", "ha": ha, "hb": hb, "batch": batch,
             "m": m, "n": n, "k": k, "dtype": dtype}
        )["h"]

    def synchronize(self) -> None:
        self._call({"op": "sync"})

    def mem_info(self):
        return tuple(self._call({"op": "mem_info"})["info"])


# dtype codes shared with _hipops
_F32, _F64 = 0, 1
_UNARY = {
    "square": 0, "negative": 1, "absolute": 2, "sqrt": 3, "exp": 4,
    "log": 5, "sin": 6, "cos": 7, "tanh": 8,
    "floor": 9, "ceil": 10, "rint": 11, "trunc": 12, "sign": 13,
    "log2": 14, "log10": 15, "exp2": 16, "expm1": 17, "log1p": 18,
    "cbrt": 19, "tan": 20, "arcsin": 21, "arccos": 22, "arctan": 23,
    "sinh": 24, "cosh": 25,
}
_BINARY = {
    "add": 0, "subtract": 1, "multiply": 2, "divide": 3, "true_divide": 3,
    "maximum": 4, "minimum": 5, "power": 6,
}
# full-array reduction modes (the extension's ReduceOp)
_REDUCE_SUM, _REDUCE_SUMSQ, _REDUCE_MAX, _REDUCE_MIN = 0, 1, 2, 3
# comparison ops (u8 mask kernels)
_CMP = {
    "less": 0, "less_equal": 1, "greater": 2, "greater_equal": 3,
    "equal": 4, "not_equal": 5,
}

MIN_ELEMS = int(os.environ.get("APP_HIP_NUMPY_MIN_ELEMS", 2_000_000))
MIN_MATMUL_FLOPS = float(os.environ.get("APP_HIP_NUMPY_MIN_MATMUL_FLOPS", 5e7))

_state = {"backend": None, "failed": None}
RPC_STATS = {"ms": 0.0, "n": 0}


def daemon_socket():
    path = os.environ.get("APP_GPU_SERVICE")
    if path and os.path.exists(path):
        return path
    return None


def daemon_configured() -> bool:
    """A GPU daemon is advertised for this engine (the socket itself may
    still be coming up -- _ensure_ready waits for it)."""
    return bool(os.environ.get("APP_GPU_SERVICE"))


def _gpu_present() -> bool:
    try:
        return _load_hipops().is_available()
    except ImportError:
        return False


def available() -> bool:
    # a CONFIGURED daemon answers availability without touching _hipops:
    # this question is asked in the ZYGOTE (numpy import hook fires during
    # preload), and probing _hipops there would load the HIP runtime
    # pre-fork -- poisoning kernel launches in every forked child (see
    # _load_hipops). Only daemon-less processes probe the local runtime.
    if daemon_configured():
        return True
    return _gpu_present()


def backend():
    _ensure_ready()
    return _state["backend"]


def _ensure_ready() -> None:
    if _state["backend"] is not None:
        return
    if _state["failed"]:
        raise RuntimeError(_state["failed"])
    try:
        env_path = os.environ.get("APP_GPU_SERVICE")
        if env_path:
            # a daemon is advertised: give it a moment to come up (its HIP
            # init runs concurrently with engine startup) before falling
            # back to an own-context backend
            import time

            wait = float(os.environ.get("APP_GPU_SERVICE_WAIT", "10"))
            deadline = time.monotonic() + wait
            while True:
                if os.path.exists(env_path):
                    try:
                        _state["backend"] = RemoteBackend(env_path)
                        return
                    except (OSError, RuntimeError):
                        pass
                if time.monotonic() >= deadline or not _gpu_present():
                    break
                time.sleep(0.05)
        _state["backend"] = LocalBackend()
    except Exception as e:
        _state["failed"] = str(e)
        raise


def warmup() -> None:
    """Connect to the GPU daemon (or bring up a local context) and touch
    the hot path once (pre-forked warm child calls this pre-request)."""
    if not available():
        raise RuntimeError("no AMD GPU visible")
    _ensure_ready()
    b = _state["backend"]
    h = b.rand(4096, _F64, 12345)
    b.sum(h, _F64, 4096, 1)
    b.free(h)


def _dtype_code(dtype):
    if dtype == _np.float64:
        return _F64
    if dtype == _np.float32:
        return _F32
    return None


class DeviceArray:
    """Device-resident array, duck-typed against numpy via NEP 13/18."""

    __array_priority__ = 1000.0

    def __init__(self, handle, shape, dtype):
        self._handle = handle
        self.shape = tuple(shape)
        self.dtype = _np.dtype(dtype)
        self._host = None  # materialized cache

    # -- basics ---------------------------------------------------------
    @property
    def size(self):
        n = 1
        for s in self.shape:
            n *= s
        return n

    @property
    def ndim(self):
        return len(self.shape)

    def __len__(self):
        if not self.shape:
            raise TypeError("len() of unsized object")
        return self.shape[0]

    def __repr__(self):
        return repr(self.materialize())

    def __str__(self):
        return str(self.materialize())

    def __del__(self):
        # free on the EXISTING backend only: a destructor running during
        # interpreter teardown (or after the backend is gone) must never
        # construct one -- _ensure_ready from GC context can re-init HIP
        try:
            b = _state["backend"]
            if b is not None and self._handle is not None:
                b.free(self._handle)
        except Exception:
            pass

    def materialize(self) -> _np.ndarray:
        if self._host is None:
            out = _np.empty(self.shape, dtype=self.dtype)
            backend().download(self._handle, out)
            self._host = out
        return self._host

    def _mutable_host(self) -> _np.ndarray:
        """Materialize for in-place mutation: the host copy becomes the
        source of truth, the (now stale) device buffer is released, and
        the next device op re-uploads lazily (`_dev_handle`). Repeated
        `x[i] = v` therefore costs one download total, not one
        upload per assignment."""
        host = self.materialize()
        if self._handle is not None:
            try:
                backend().free(self._handle)
            except Exception:
                pass
            self._handle = None
        return host

    def _dev_handle(self):
        if self._handle is None:
            arr = _np.ascontiguousarray(self.materialize())
            self._handle = backend().upload(arr)
        return self._handle

    def __array__(self, dtype=None, copy=None):
        host = self.materialize()
        if dtype is not None and dtype != host.dtype:
            return host.astype(dtype)
        return host

    def astype(self, dtype, **kwargs):
        """Device-side cast between f32/f64 (a common follow-on to the
        routed RNG entry points); anything else materializes."""
        target = _np.dtype(dtype)
        src = _dtype_code(self.dtype)
        dst = _dtype_code(target)
        if src is not None and dst is not None and not kwargs:
            if src == dst:
                return self
            h = backend().convert(self._dev_handle(), src, dst, self.size)
            return DeviceArray(h, self.shape, target)
        return self.materialize().astype(dtype, **kwargs)

    # anything we don't implement: materialize and delegate
    def __getattr__(self, name):
        return getattr(self.materialize(), name)

    # scalar reads of big resident arrays fetch one element, not the
    # whole buffer (x[i] in numpy returns a copy -- a scalar -- so this
    # fast path is semantically exact)
    _SCALAR_FETCH_MIN = 1 << 20  # elements; below this just materialize

    def __getitem__(self, idx):
        if (
            self._host is None
            and self._handle is not None
            and self.size >= self._SCALAR_FETCH_MIN
            and self.dtype.itemsize in (4, 8)
        ):
            flat = self._flat_index(idx)
            if flat is not None:
                raw = backend().download_slice(
                    self._handle, flat * self.dtype.itemsize,
                    self.dtype.itemsize,
                )
                return _np.frombuffer(raw, dtype=self.dtype)[0]
        return self.materialize()[idx]

    def _flat_index(self, idx):
        """C-contiguous flat offset for an all-integer index, or None."""
        if isinstance(idx, (int, _np.integer)):
            idx = (idx,)
        if not (
            isinstance(idx, tuple)
            and len(idx) == len(self.shape)
            and all(isinstance(i, (int, _np.integer)) for i in idx)
        ):
            return None
        flat = 0
        for i, dim in zip(idx, self.shape):
            i = int(i)
            if i < 0:
                i += dim
            if not 0 <= i < dim:
                return None  # let numpy raise its own IndexError
            flat = flat * dim + i
        return flat

    def __setitem__(self, idx, value):
        # in-place mutation (x[0] = 1, x[x < 0] = 0, ...): CPU numpy
        # supports it, so the sandbox contract requires it. Boolean-mask
        # assignment of a scalar runs ON DEVICE (masked_fill on this
        # array's buffer -- safe, handles are single-owner); everything
        # else materializes, mutates the host copy, and drops the device
        # buffer (re-uploaded lazily on the next device op).
        if (
            isinstance(value, (int, float))
            and not isinstance(value, bool)
            and _dtype_code(self.dtype) is not None
            and self._handle is not None
        ):
            mask = None
            if isinstance(idx, BoolDeviceArray) and idx.shape == self.shape:
                mask = idx
            elif (
                isinstance(idx, _np.ndarray)
                and idx.dtype == _np.bool_
                and idx.shape == self.shape
            ):
                mh = backend().upload(_np.ascontiguousarray(idx))
                mask = BoolDeviceArray(mh, idx.shape, _np.bool_)
            if mask is not None:
                backend().masked_fill(
                    self._handle, mask._dev_handle(),
                    _dtype_code(self.dtype), self.size, float(value),
                )
                self._host = None  # the device buffer is authoritative
                return
        if isinstance(idx, DeviceArray):
            idx = idx.materialize()
        if isinstance(value, DeviceArray):
            value = value.materialize()
        self._mutable_host()[idx] = value

    def __float__(self):
        return float(self.materialize())

    def __bool__(self):
        # numpy semantics: truthiness only for size-1 arrays, else raise
        return bool(self.materialize())

    def __iter__(self):
        return iter(self.materialize())

    # comparisons produce DEVICE boolean masks (u8 buffers) for scalar
    # and same-shape operands; anything else materializes and delegates
    # (the default object identity would silently return False for
    # `x == 5`-style masks)
    def _compare(self, cmp_name, other):
        code = _dtype_code(self.dtype)
        if code is None:
            return None
        if isinstance(other, (bool, _np.bool_)):
            return None
        if isinstance(other, (int, float)):
            h = backend().compare(
                self._dev_handle(), code, self.size, _CMP[cmp_name], 0,
                float(other),
            )
        elif (
            isinstance(other, DeviceArray)
            and other.shape == self.shape
            and other.dtype == self.dtype
        ):
            h = backend().compare(
                self._dev_handle(), code, self.size, _CMP[cmp_name],
                other._dev_handle(), 0.0,
            )
        else:
            return None
        return BoolDeviceArray(h, self.shape, _np.bool_)

    def __eq__(self, o):
        r = self._compare("equal", o)
        return self.materialize() == _asarray_or_scalar(o) if r is None else r

    def __ne__(self, o):
        r = self._compare("not_equal", o)
        return self.materialize() != _asarray_or_scalar(o) if r is None else r

    def __lt__(self, o):
        r = self._compare("less", o)
        return self.materialize() < _asarray_or_scalar(o) if r is None else r

    def __le__(self, o):
        r = self._compare("less_equal", o)
        return self.materialize() <= _asarray_or_scalar(o) if r is None else r

    def __gt__(self, o):
        r = self._compare("greater", o)
        return self.materialize() > _asarray_or_scalar(o) if r is None else r

    def __ge__(self, o):
        r = self._compare("greater_equal", o)
        return self.materialize() >= _asarray_or_scalar(o) if r is None else r

    __hash__ = None  # mutable-array semantics, same as numpy

    # -- device compute -------------------------------------------------
    def _unary(self, opname):
        out = backend().unary(self._dev_handle(), _UNARY[opname], _dtype_code(self.dtype), self.size)
        return DeviceArray(out, self.shape, self.dtype)

    def _binary(self, opname, other, reverse=False):
        code = _dtype_code(self.dtype)
        if isinstance(other, DeviceArray):
            if other.shape != self.shape or other.dtype != self.dtype:
                if not reverse:
                    r = self._binary_bcast(opname, other)
                    if r is not NotImplemented:
                        return r
                return NotImplemented
            a, b = (other, self) if reverse else (self, other)
            out = backend().binary(a._dev_handle(), b._dev_handle(), _BINARY[opname], code, self.size)
            return DeviceArray(out, self.shape, self.dtype)
        if isinstance(other, (int, float)):
            if reverse and opname in ("subtract", "divide", "true_divide", "power"):
                return NotImplemented  # scalar-first sub/div/pow: fall back
            out = backend().binary_scalar(
                self._dev_handle(), float(other), _BINARY[opname], code, self.size
            )
            return DeviceArray(out, self.shape, self.dtype)
        return NotImplemented

    def _binary_bcast(self, opname, other):
        """Device broadcasting for the two clean numpy cases:
        row vector (other.shape == self.shape[-1:], broadcast along the
        leading axes) and keepdims column (other.shape == self.shape with
        the last axis == 1). Returns NotImplemented otherwise."""
        if not isinstance(other, DeviceArray) or other.dtype != self.dtype:
            return NotImplemented
        code = _dtype_code(self.dtype)
        if code is None or len(self.shape) < 2:
            return NotImplemented
        inner = self.shape[-1]
        outer = self.size // max(1, inner)
        if other.shape == self.shape[-1:]:
            mode = 0  # b[inner] along outer
        elif other.shape == self.shape[:-1] + (1,):
            mode = 1  # b[outer] along inner
        else:
            return NotImplemented
        out = backend().binary_bcast(
            self._dev_handle(), other._dev_handle(), _BINARY[opname], code,
            outer, inner, mode,
        )
        return DeviceArray(out, self.shape, self.dtype)

    @staticmethod
    def _norm_axis(axis, nd):
        if isinstance(axis, (tuple, list)):
            if len(axis) != 1:
                return None
            axis = axis[0]
        if not isinstance(axis, (int, _np.integer)):
            return None
        axis = int(axis)
        if axis < 0:
            axis += nd
        return axis if 0 <= axis < nd else None

    def _axis_reduce(self, mode, axis, keepdims=False):
        """Single-axis reduction on-device: the contiguous array viewed
        as [outer][red][inner], reduced over the middle. Returns None
        when not routable (caller falls back to host numpy)."""
        code = _dtype_code(self.dtype)
        if code is None:
            return None
        nd = len(self.shape)
        axis = self._norm_axis(axis, nd)
        if axis is None or self.size == 0:
            return None
        outer = 1
        for sdim in self.shape[:axis]:
            outer *= sdim
        red = self.shape[axis]
        inner = 1
        for sdim in self.shape[axis + 1:]:
            inner *= sdim
        if inner == 1 and outer < 64:
            # last-axis reduce with very few slices: the wave-per-slice
            # kernel would use < 64 of the chip's ~8k wave slots -- let
            # the caller pick a better path (full-reduce or host)
            return None
        h = backend().reduce_axis(
            self._dev_handle(), code, outer, red, inner, mode
        )
        out_shape = (
            self.shape[:axis]
            + ((1,) if keepdims else ())
            + self.shape[axis + 1:]
        )
        return DeviceArray(h, out_shape, self.dtype)

    def _full_axis_scalar(self, axis, kwargs, compute):
        """axis reduce of a 1-D array == full reduce: route to the
        scalar kernel (numpy returns a scalar, or a 1-element array
        with keepdims)."""
        if (
            len(self.shape) == 1
            and set(kwargs) <= {"keepdims"}
            and self._norm_axis(axis, 1) == 0
        ):
            val = compute()
            return _np.array([val]) if kwargs.get("keepdims") else val
        return None

    def sum(self, axis=None, **kwargs):
        if axis is None and not kwargs.get("keepdims"):
            return self.dtype.type(
                backend().sum(self._dev_handle(), _dtype_code(self.dtype), self.size, 0)
            )
        if axis is not None:
            r = self._full_axis_scalar(axis, kwargs, lambda: self.sum())
            if r is not None:
                return r
        if axis is not None and set(kwargs) <= {"keepdims"}:
            r = self._axis_reduce(_REDUCE_SUM, axis, kwargs.get("keepdims", False))
            if r is not None:
                return r
        return self.materialize().sum(axis=axis, **kwargs)

    def mean(self, axis=None, **kwargs):
        if axis is None and not kwargs.get("keepdims"):
            return self.dtype.type(float(self.sum()) / self.size)
        if axis is not None:
            r = self._full_axis_scalar(axis, kwargs, lambda: self.mean())
            if r is not None:
                return r
        if axis is not None and set(kwargs) <= {"keepdims"}:
            r = self._axis_reduce(_REDUCE_SUM, axis, kwargs.get("keepdims", False))
            if r is not None:
                nd = len(self.shape)
                ax = axis if not isinstance(axis, (tuple, list)) else axis[0]
                red = self.shape[int(ax) if int(ax) >= 0 else int(ax) + nd]
                out = r._binary("multiply", 1.0 / red)
                if out is not NotImplemented:
                    return out
        return self.materialize().mean(axis=axis, **kwargs)

    def square_sum(self):
        """Fused sum(x*x) -- no intermediate array."""
        return self.dtype.type(
            backend().sum(self._dev_handle(), _dtype_code(self.dtype), self.size, 1)
        )

    def max(self, axis=None, **kwargs):
        if axis is None and not kwargs.get("keepdims"):
            return self.dtype.type(
                backend().sum(
                    self._dev_handle(), _dtype_code(self.dtype), self.size, _REDUCE_MAX
                )
            )
        if axis is not None:
            r = self._full_axis_scalar(axis, kwargs, lambda: self.max())
            if r is not None:
                return r
        if axis is not None and set(kwargs) <= {"keepdims"}:
            r = self._axis_reduce(_REDUCE_MAX, axis, kwargs.get("keepdims", False))
            if r is not None:
                return r
        return self.materialize().max(axis=axis, **kwargs)

    def min(self, axis=None, **kwargs):
        if axis is None and not kwargs.get("keepdims"):
            return self.dtype.type(
                backend().sum(
                    self._dev_handle(), _dtype_code(self.dtype), self.size, _REDUCE_MIN
                )
            )
        if axis is not None:
            r = self._full_axis_scalar(axis, kwargs, lambda: self.min())
            if r is not None:
                return r
        if axis is not None and set(kwargs) <= {"keepdims"}:
            r = self._axis_reduce(_REDUCE_MIN, axis, kwargs.get("keepdims", False))
            if r is not None:
                return r
        return self.materialize().min(axis=axis, **kwargs)

    def clip(self, a_min=None, a_max=None, **kwargs):
        """Scalar-bounds clip on device (maximum then minimum chains);
        array bounds / extra kwargs fall back to host numpy."""
        if not kwargs and (
            a_min is None or isinstance(a_min, (int, float))
        ) and (a_max is None or isinstance(a_max, (int, float))):
            r = self
            if a_min is not None:
                r2 = r._binary("maximum", float(a_min))
                if r2 is NotImplemented:
                    return self.materialize().clip(a_min, a_max, **kwargs)
                r = r2
            if a_max is not None:
                r2 = r._binary("minimum", float(a_max))
                if r2 is NotImplemented:
                    return self.materialize().clip(a_min, a_max, **kwargs)
                r = r2
            if r is not self:
                return r
        return self.materialize().clip(a_min, a_max, **kwargs)

    def cumsum(self, axis=None, **kwargs):
        """Cumulative sum on device: flat (axis=None / 1-D), or 2-D
        along either axis (axis=0 through the device transpose);
        accumulation in double. Rounding may differ from numpy's
        strictly-sequential order by ~1 ulp at block regroupings."""
        code = _dtype_code(self.dtype)
        if code is not None and not kwargs and self.size > 0:
            if axis is None or (
                len(self.shape) == 1 and self._norm_axis(axis, 1) == 0
            ):
                h = backend().cumsum(self._dev_handle(), code, self.size)
                shape = self.shape if axis is not None else (self.size,)
                return DeviceArray(h, shape, self.dtype)
            if len(self.shape) == 2 and axis is not None:
                ax = self._norm_axis(axis, 2)
                if ax == 1:
                    rows, cols = self.shape
                    h = backend().cumsum2d(
                        self._dev_handle(), code, rows, cols
                    )
                    return DeviceArray(h, self.shape, self.dtype)
                if ax == 0:
                    t = self._device_transposed()
                    return t.cumsum(axis=1)._device_transposed()
        return self.materialize().cumsum(axis=axis, **kwargs)

    def _device_clone(self):
        """Device-side copy (one ~5 TB/s pass; a fused add-0), vs the
        ~15 GB/s host roundtrip a materialize would cost."""
        code = _dtype_code(self.dtype)
        h = backend().binary_scalar(
            self._dev_handle(), 0.0, _BINARY["add"], code, self.size
        )
        return h

    def reshape(self, *shape, **kwargs):
        """Device reshape: C-contiguous only, returns a device COPY
        (numpy returns a view where possible -- write-through to the
        parent is the one divergence; reads are identical)."""
        if len(shape) == 1 and isinstance(shape[0], (tuple, list)):
            shape = tuple(shape[0])
        ok = (
            _dtype_code(self.dtype) is not None
            and self._host is None
            and kwargs.get("order", "C") == "C"
            and set(kwargs) <= {"order"}
        )
        if ok:
            dims = [int(d) for d in shape]
            neg = [i for i, d in enumerate(dims) if d < 0]
            if len(neg) <= 1:
                known = 1
                for d in dims:
                    if d >= 0:
                        known *= d
                if neg:
                    if known > 0 and self.size % known == 0:
                        dims[neg[0]] = self.size // known
                    else:
                        ok = False
                if ok and _np.prod(dims, dtype=_np.int64) == self.size:
                    return DeviceArray(
                        self._device_clone(), tuple(dims), self.dtype
                    )
        return self.materialize().reshape(*shape, **kwargs)

    def ravel(self, order="C"):
        if order == "C" and _dtype_code(self.dtype) is not None \
                and self._host is None:
            return DeviceArray(self._device_clone(), (self.size,), self.dtype)
        return self.materialize().ravel(order)

    def flatten(self, order="C"):
        return self.ravel(order)

    def _device_transposed(self):
        """Device 2-D transpose (LDS-tiled). Itemsize-based: also moves
        int64 index matrices through the f64-width kernel (pure data
        movement)."""
        rows, cols = self.shape
        code = _F64 if self.dtype.itemsize == 8 else _F32
        h = backend().transpose(self._dev_handle(), code, rows, cols)
        return DeviceArray(h, (cols, rows), self.dtype)

    def transpose(self, *axes):
        if len(self.shape) == 2 and self.dtype.itemsize in (4, 8) and (
            not axes or axes == (1, 0) or axes == ((1, 0),)
        ):
            return self._device_transposed()
        return self.materialize().transpose(*axes)

    @property
    def T(self):
        if len(self.shape) < 2:
            return self
        return self.transpose()

    def _sort_routable(self, axis, kind, order, kwargs):
        """np.sort/argsort route: f32/f64, default comparator, flat 1-D
        or 2-D along either axis (axis=0 goes through the device
        transpose). The device sort is an LSD radix sort, so it is
        stable -- every numpy `kind` is satisfied."""
        if (
            _dtype_code(self.dtype) is None
            or self.size == 0
            or kwargs
            or order is not None
            or kind not in (None, "stable", "quicksort", "mergesort",
                            "heapsort")
        ):
            return False
        if len(self.shape) == 1:
            return axis is None or self._norm_axis(axis, 1) == 0
        if len(self.shape) == 2 and axis is not None:
            return (
                self._norm_axis(axis, 2) in (0, 1)
                and self.size <= (1 << 31)
            )
        return False

    def _device_sorted(self, want_idx, axis=None):
        """axis=None: flat (1-D) or per-row (2-D). axis=0 on 2-D:
        transpose -> row sort -> transpose back."""
        code = _dtype_code(self.dtype)
        if len(self.shape) == 2 and axis == 0:
            t = self._device_transposed()
            if want_idx:
                sv, si = t._device_sorted(True)
                return sv._device_transposed(), si._device_transposed()
            return t._device_sorted(False)._device_transposed()
        if len(self.shape) == 2:
            rows, cols = self.shape
            r = backend().sort2d(
                self._dev_handle(), code, rows, cols, 1 if want_idx else 0
            )
        else:
            r = backend().sort(
                self._dev_handle(), code, self.size, 1 if want_idx else 0
            )
        if want_idx:
            h, hi = r
            return (
                DeviceArray(h, self.shape, self.dtype),
                DeviceArray(hi, self.shape, _np.int64),
            )
        return DeviceArray(r, self.shape, self.dtype)

    def _sort_axis01(self, axis):
        if len(self.shape) == 2 and axis is not None:
            return self._norm_axis(axis, 2)
        return None

    def sort(self, axis=-1, kind=None, order=None, **kwargs):
        """In-place sort (ndarray.sort contract): the handle is swapped
        for the sorted buffer."""
        if self._sort_routable(axis, kind, order, kwargs):
            res = self._device_sorted(False, axis=self._sort_axis01(axis))
            old_h, self._handle = self._handle, res._handle
            res._handle = None  # ownership moved; res.__del__ must not free
            self._host = None
            if old_h is not None:
                try:
                    backend().free(old_h)
                except Exception:
                    pass
            return None
        host = self._mutable_host()
        host.sort(axis=axis, kind=kind, order=order, **kwargs)
        return None

    def argsort(self, axis=-1, kind=None, order=None, **kwargs):
        if self._sort_routable(axis, kind, order, kwargs):
            _, idx = self._device_sorted(True, axis=self._sort_axis01(axis))
            return idx
        return self.materialize().argsort(
            axis=axis, kind=kind, order=order, **kwargs
        )

    def partition(self, kth, axis=-1, kind="introselect", order=None,
                  **kwargs):
        """In-place partition. A fully sorted array satisfies the
        partition contract for every kth, so the device route reuses
        the radix sort."""
        if self._sort_routable(axis, None, order, kwargs):
            self.sort(axis=axis)
            return None
        host = self._mutable_host()
        host.partition(kth, axis=axis, kind=kind, order=order, **kwargs)
        return None

    def argpartition(self, kth, axis=-1, kind="introselect", order=None,
                     **kwargs):
        if self._sort_routable(axis, None, order, kwargs):
            return self.argsort(axis=axis)
        return self.materialize().argpartition(
            kth, axis=axis, kind=kind, order=order, **kwargs
        )

    def round(self, decimals=0, **kwargs):
        if decimals == 0 and not kwargs and _dtype_code(self.dtype) is not None:
            return self._unary("rint")
        return self.materialize().round(decimals, **kwargs)

    def isnan(self):
        """Device NaN mask (x != x elementwise)."""
        r = self._compare("not_equal", self)
        if r is None:
            return _np.isnan(self.materialize())
        return r

    def count_nonzero(self):
        m = self._compare("not_equal", 0.0)
        if m is None:
            return _np.count_nonzero(self.materialize())
        return m.sum()

    def any(self, axis=None, **kwargs):
        if axis is None and not kwargs and _dtype_code(self.dtype) is not None:
            return bool(int(self.count_nonzero()) > 0)
        return self.materialize().any(axis=axis, **kwargs)

    def all(self, axis=None, **kwargs):
        if axis is None and not kwargs and _dtype_code(self.dtype) is not None:
            return bool(int(self.count_nonzero()) == self.size)
        return self.materialize().all(axis=axis, **kwargs)

    def argmax(self, axis=None, **kwargs):
        if axis is None and not kwargs and _dtype_code(self.dtype) is not None:
            return _np.intp(
                backend().argminmax(
                    self._dev_handle(), _dtype_code(self.dtype), self.size, 1
                )
            )
        return self.materialize().argmax(axis=axis, **kwargs)

    def argmin(self, axis=None, **kwargs):
        if axis is None and not kwargs and _dtype_code(self.dtype) is not None:
            return _np.intp(
                backend().argminmax(
                    self._dev_handle(), _dtype_code(self.dtype), self.size, 0
                )
            )
        return self.materialize().argmin(axis=axis, **kwargs)

    def var(self, axis=None, ddof=0, **kwargs):
        """Two-pass variance entirely on-device: mean, then the fused
        sum((x-mean)^2). (The one-pass sum-of-squares form cancels
        catastrophically when |mean| >> std and can go negative.)
        2-D axis-wise variance composes the same way from the axis
        reducers + broadcast subtract."""
        if axis is None and not kwargs.get("keepdims"):
            n = self.size
            mu = float(self.sum()) / n
            shifted = self._binary("subtract", mu)
            if shifted is NotImplemented:
                return self.materialize().var(ddof=ddof)
            return self.dtype.type(float(shifted.square_sum()) / (n - ddof))
        if (
            axis is not None
            and len(self.shape) == 2
            and not kwargs.get("keepdims")
            and set(kwargs) <= {"keepdims"}
            and _dtype_code(self.dtype) is not None
        ):
            ax = self._norm_axis(axis, 2)
            if ax is not None:
                red = self.shape[ax]
                mu = self.mean(axis=ax, keepdims=(ax == 1))
                if isinstance(mu, DeviceArray):
                    centered = self._binary_bcast("subtract", mu)
                    if centered is not NotImplemented:
                        sq = centered._unary("square")
                        total = sq.sum(axis=ax)
                        if isinstance(total, DeviceArray):
                            r = total._binary(
                                "multiply", 1.0 / (red - ddof)
                            )
                            if r is not NotImplemented:
                                return r
        return self.materialize().var(axis=axis, ddof=ddof, **kwargs)

    def std(self, axis=None, ddof=0, **kwargs):
        if axis is None and not kwargs.get("keepdims"):
            return self.dtype.type(float(self.var(ddof=ddof)) ** 0.5)
        if axis is not None and not kwargs.get("keepdims"):
            v = self.var(axis=axis, ddof=ddof, **kwargs)
            if isinstance(v, DeviceArray):
                return v._unary("sqrt")
        return self.materialize().std(axis=axis, ddof=ddof, **kwargs)

    # -- NEP 13: ufuncs --------------------------------------------------
    def __array_ufunc__(self, ufunc, method, *inputs, **kwargs):
        if kwargs.get("out") is not None:
            return self._fallback_ufunc(ufunc, method, inputs, kwargs)
        name = ufunc.__name__
        if method == "__call__":
            if name == "isnan" and len(inputs) == 1 and inputs[0] is self:
                if isinstance(self, BoolDeviceArray):
                    return _np.zeros(self.shape, dtype=bool)
                return self.isnan()
            if name in (
                "logical_and", "logical_or", "logical_xor",
                "bitwise_and", "bitwise_or", "bitwise_xor",
            ) and len(inputs) == 2 and isinstance(self, BoolDeviceArray):
                other = inputs[1] if inputs[0] is self else inputs[0]
                lop = {
                    "logical_and": 0, "bitwise_and": 0,
                    "logical_or": 1, "bitwise_or": 1,
                    "logical_xor": 2, "bitwise_xor": 2,
                }[name]
                r = self._logic(other, lop)
                if r is not NotImplemented:
                    return r
            if name == "logical_not" and len(inputs) == 1 and isinstance(
                self, BoolDeviceArray
            ):
                return self._logic(None, 4)
            if name == "matmul" and len(inputs) == 2:
                r = matmul(inputs[0], inputs[1], _force=True)
                if r is not NotImplemented:
                    return r
            if name in _CMP and len(inputs) == 2 and inputs[0] is self:
                r = self._compare(name, inputs[1])
                if r is not None:
                    return r
            if name in _UNARY and len(inputs) == 1 and inputs[0] is self \
                    and _dtype_code(self.dtype) is not None:
                return self._unary(name)
            if name in _BINARY and len(inputs) == 2:
                a, b = inputs
                if a is self:
                    r = self._binary(name, b)
                elif b is self:
                    r = self._binary(name, a, reverse=True)
                else:
                    r = NotImplemented
                if r is not NotImplemented:
                    return r
        elif method == "reduce" and len(inputs) == 1:
            axis = kwargs.get("axis")
            keepdims = bool(kwargs.get("keepdims", False))
            meth = {"add": "sum", "maximum": "max", "minimum": "min"}.get(name)
            if meth and set(kwargs) <= {"axis", "keepdims"}:
                if axis is None and not keepdims:
                    return getattr(inputs[0], meth)()
                if axis is not None:
                    return getattr(inputs[0], meth)(axis=axis, keepdims=keepdims)
        return self._fallback_ufunc(ufunc, method, inputs, kwargs)

    def _fallback_ufunc(self, ufunc, method, inputs, kwargs):
        host_inputs = [
            x.materialize() if isinstance(x, DeviceArray) else x for x in inputs
        ]
        out = kwargs.get("out")
        if out is not None:
            # numpy rejects duck-typed out= targets: substitute each
            # DeviceArray's mutable host copy (updated in place, device
            # buffer invalidated) and hand the originals back
            outs = out if isinstance(out, tuple) else (out,)
            host_outs = tuple(
                o._mutable_host() if isinstance(o, DeviceArray) else o
                for o in outs
            )
            kwargs = {**kwargs, "out": host_outs}
            result = getattr(ufunc, method)(*host_inputs, **kwargs)
            if isinstance(result, tuple):
                return tuple(
                    orig if isinstance(orig, DeviceArray) else res
                    for orig, res in zip(outs, result)
                )
            return outs[0] if isinstance(outs[0], DeviceArray) else result
        return getattr(ufunc, method)(*host_inputs, **kwargs)

    # -- NEP 18: numpy functions -----------------------------------------
    def __array_function__(self, func, types, args, kwargs):
        for stage in (_af_linalg, _af_order_stats, _af_array_ops, _af_cleanup, _af_structure):
            r = stage(func, args, kwargs)
            if r is not _AF_PASS:
                return r
        # generic fallback: materialize every DeviceArray
        host_args = [
            x.materialize() if isinstance(x, DeviceArray) else x for x in args
        ]
        return func(*host_args, **kwargs)

    # -- operators: device kernel when routable, otherwise numpy's own
    # semantics on the host copies (dtype upcasts, broadcasting beyond
    # the device cases, bool operands, ... must all behave exactly like
    # CPU numpy -- raising here would break valid user code)
    def _op_or_host(self, opname, o, host_op, reverse=False):
        r = self._binary(opname, o, reverse=reverse)
        if r is not NotImplemented:
            return r
        a = self.materialize()
        b = _asarray_or_scalar(o)
        return host_op(b, a) if reverse else host_op(a, b)

    def __add__(self, o):
        return self._op_or_host("add", o, lambda a, b: a + b)

    def __radd__(self, o):
        return self._op_or_host("add", o, lambda a, b: a + b, reverse=True)

    def __sub__(self, o):
        return self._op_or_host("subtract", o, lambda a, b: a - b)

    def __mul__(self, o):
        return self._op_or_host("multiply", o, lambda a, b: a * b)

    def __rmul__(self, o):
        return self._op_or_host("multiply", o, lambda a, b: a * b, reverse=True)

    def __truediv__(self, o):
        return self._op_or_host("divide", o, lambda a, b: a / b)

    def __pow__(self, o):
        return self._op_or_host("power", o, lambda a, b: a ** b)

    # scalar-first sub/div/pow have no device kernel ordering: compute on
    # the host instead of raising (user code does `1.0 / x` freely)
    def __rsub__(self, o):
        return o - self.materialize()

    def __rtruediv__(self, o):
        return o / self.materialize()

    def __rpow__(self, o):
        return o ** self.materialize()

    def __matmul__(self, o):
        r = matmul(self, o, _force=True)
        return self.materialize() @ _asarray(o) if r is NotImplemented else r




class BoolDeviceArray(DeviceArray):
    """Device-resident boolean mask (u8 storage, numpy bool semantics).
    Arithmetic/reduction kernels are f32/f64-only, so everything except
    the mask-specific fast paths (popcount sum, np.where selection,
    masked assignment) materializes to a host bool array."""

    def sum(self, axis=None, **kwargs):
        if axis is None and not kwargs:
            return _np.intp(backend().mask_count(self._dev_handle(), self.size))
        r = self._axis_count(axis, kwargs)
        if r is not None:
            return r
        return self.materialize().sum(axis=axis, **kwargs)

    def _axis_count(self, axis, kwargs):
        """Per-axis True counts ((x > 0).sum(axis=1)): mask -> 0/1 f64
        via the select kernel, axis reduce, downloaded as numpy's int64
        (the result is outer-size small)."""
        if kwargs or self.size == 0:
            return None
        ones = where_device(self, 1.0, 0.0)
        if ones is NotImplemented:
            return None
        r = ones._axis_reduce(_REDUCE_SUM, axis)
        if r is None:
            return None
        return r.materialize().astype(_np.int64)

    def count_nonzero(self):
        return self.sum()

    def mean(self, axis=None, **kwargs):
        if axis is None and not kwargs:
            return _np.float64(int(self.sum()) / self.size)
        r = self._axis_count(axis, kwargs)
        if r is not None:
            red = self.shape[self._norm_axis(axis, len(self.shape))]
            return r / float(red)
        return self.materialize().mean(axis=axis, **kwargs)

    def any(self, axis=None, **kwargs):
        if axis is None and not kwargs:
            return bool(int(self.sum()) > 0)
        r = self._axis_count(axis, kwargs)
        if r is not None:
            return r > 0
        return self.materialize().any(axis=axis, **kwargs)

    def all(self, axis=None, **kwargs):
        if axis is None and not kwargs:
            return bool(int(self.sum()) == self.size)
        r = self._axis_count(axis, kwargs)
        if r is not None:
            red = self.shape[self._norm_axis(axis, len(self.shape))]
            return r == red
        return self.materialize().all(axis=axis, **kwargs)

    # f32/f64-only device paths must not see a u8 buffer
    def _unary(self, opname):
        raise TypeError("unary op on boolean mask")

    def _binary(self, opname, other, reverse=False):
        return NotImplemented

    def max(self, axis=None, **kwargs):
        return self.materialize().max(axis=axis, **kwargs)

    def min(self, axis=None, **kwargs):
        return self.materialize().min(axis=axis, **kwargs)

    def argmax(self, axis=None, **kwargs):
        return self.materialize().argmax(axis=axis, **kwargs)

    def argmin(self, axis=None, **kwargs):
        return self.materialize().argmin(axis=axis, **kwargs)

    def var(self, axis=None, ddof=0, **kwargs):
        return self.materialize().var(axis=axis, ddof=ddof, **kwargs)

    def std(self, axis=None, ddof=0, **kwargs):
        return self.materialize().std(axis=axis, ddof=ddof, **kwargs)

    def astype(self, dtype, **kwargs):
        return self.materialize().astype(dtype, **kwargs)

    # mask logic stays on device: (x > 0) & (x < 1), ~mask, mask | other
    def _logic(self, other, lop):
        if other is None:
            h = backend().mask_logic(self._dev_handle(), 0, self.size, lop)
        elif (
            isinstance(other, BoolDeviceArray) and other.shape == self.shape
        ):
            h = backend().mask_logic(
                self._dev_handle(), other._dev_handle(), self.size, lop
            )
        else:
            return NotImplemented
        return BoolDeviceArray(h, self.shape, _np.bool_)

    def __and__(self, o):
        r = self._logic(o, 0)
        if r is NotImplemented:
            return self.materialize() & _asarray_or_scalar(o)
        return r

    def __or__(self, o):
        r = self._logic(o, 1)
        if r is NotImplemented:
            return self.materialize() | _asarray_or_scalar(o)
        return r

    def __xor__(self, o):
        r = self._logic(o, 2)
        if r is NotImplemented:
            return self.materialize() ^ _asarray_or_scalar(o)
        return r

    def __invert__(self):
        return self._logic(None, 4)

    __rand__ = __and__
    __ror__ = __or__
    __rxor__ = __xor__


_QUANTILE_BINS = 4096
_QUANTILE_EXTRACT_CAP = 1 << 20  # 8 MB of f64 candidates, max


def _order_stat_device(x, k):
    """k-th (0-based) order statistic of a DeviceArray via histogram
    bisection + a final exact extraction. Returns (value, nan_count)."""
    code = _dtype_code(x.dtype)
    n = x.size
    lo = float(x.min())
    hi = float(x.max())
    if lo != lo or hi != hi:  # min/max NaN-propagate: NaNs present
        return float("nan"), 1

    below = 0
    for _ in range(64):
        if not hi > lo:
            return lo, 0
        counts = _np.frombuffer(
            backend().histogram(
                x._dev_handle(), code, n, lo, hi, _QUANTILE_BINS
            ),
            dtype=_np.uint64,
        )
        nan_count = int(counts[_QUANTILE_BINS])
        if nan_count:
            return float("nan"), nan_count
        width = (hi - lo) / _QUANTILE_BINS
        cs = counts[:_QUANTILE_BINS].astype(_np.int64).cumsum()
        rel_k = k - below
        target_bin = int(_np.searchsorted(cs, rel_k, side="right"))
        if target_bin >= _QUANTILE_BINS:
            return hi, 0  # k at the far edge (fp rounding): max wins
        cum = below + (int(cs[target_bin - 1]) if target_bin else 0)
        in_bin = int(counts[target_bin])
        new_lo = lo + target_bin * width
        new_hi = (
            lo + (target_bin + 1) * width
            if target_bin + 1 < _QUANTILE_BINS
            else hi
        )
        if in_bin <= _QUANTILE_EXTRACT_CAP:
            _, data = backend().extract_range(
                x._dev_handle(), code, n, new_lo, new_hi,
                _QUANTILE_EXTRACT_CAP,
            )
            vals = _np.sort(_np.frombuffer(data, dtype=_np.float64))
            if len(vals) == 0:
                return new_lo, 0
            idx = min(max(k - cum, 0), len(vals) - 1)
            return float(vals[idx]), 0
        below = cum
        lo, hi = new_lo, new_hi
    return float("nan"), 0  # did not converge: caller falls back


def quantile_rows_device_multi(x, qs):
    """Per-row quantiles for an ARRAY of q's: ONE row sort, then two
    pitched column downloads per q. Returns [len(qs)][rows] like numpy
    (quantiles axis leads)."""
    if not isinstance(x, DeviceArray) or len(x.shape) != 2:
        return None
    if _dtype_code(x.dtype) is None:
        return None
    rows, cols = x.shape
    if cols < 1 or rows < 1 or x.size > (1 << 31):
        return None
    srt = x._device_sorted(False)
    esz = x.dtype.itemsize

    def col(k):
        raw = backend().download_strided(
            srt._dev_handle(), k * esz, cols * esz, esz, rows
        )
        return _np.frombuffer(raw, dtype=x.dtype).copy()

    top = col(cols - 1)
    nan_rows = _np.isnan(top)
    out = _np.empty((len(qs), rows), dtype=x.dtype)
    cache = {}
    for i, q in enumerate(qs):
        if not 0.0 <= float(q) <= 1.0:
            return None
        pos = float(q) * (cols - 1)
        k0 = int(_np.floor(pos))
        k1 = min(k0 + 1, cols - 1)
        frac = x.dtype.type(pos - k0)
        if k0 not in cache:
            cache[k0] = col(k0)
        c0 = cache[k0]
        if k1 == k0 or frac == 0:
            row = c0.copy()
        else:
            if k1 not in cache:
                cache[k1] = col(k1)
            row = c0 + (cache[k1] - c0) * frac
        row[nan_rows] = _np.nan
        out[i] = row
    return out


def quantile_rows_device(x, q):
    """Per-row quantile (axis=-1) of a 2-D DeviceArray: device row sort,
    then ONE pitched column download per interpolation endpoint (R
    elements, not R*L). numpy-linear interpolation; rows containing NaN
    yield NaN (numpy parity, minus numpy's RuntimeWarning). Returns a
    host ndarray of length R, or None when not routable."""
    if not isinstance(x, DeviceArray) or len(x.shape) != 2:
        return None
    if _dtype_code(x.dtype) is None or not 0.0 <= q <= 1.0:
        return None
    rows, cols = x.shape
    if cols < 1 or rows < 1 or x.size > (1 << 31):
        return None
    srt = x._device_sorted(False)
    esz = x.dtype.itemsize

    def col(k):
        raw = backend().download_strided(
            srt._dev_handle(), k * esz, cols * esz, esz, rows
        )
        return _np.frombuffer(raw, dtype=x.dtype).copy()

    pos = q * (cols - 1)
    k0 = int(_np.floor(pos))
    k1 = min(k0 + 1, cols - 1)
    frac = x.dtype.type(pos - k0)
    c0 = col(k0)
    out = c0 if k1 == k0 or frac == 0 else c0 + (col(k1) - c0) * frac
    top = col(cols - 1)  # NaNs sort last: top column flags NaN rows
    out[_np.isnan(top)] = _np.nan
    return out


def nan_reduce_device(x, kind, ddof=0):
    """Flat NaN-ignoring reductions (np.nansum/nanmean/nanmax/nanmin/
    nanstd/nanvar) composed from the device mask ops: isnan mask ->
    popcount -> where-replace -> fused reduce. Returns None when not
    routable (caller falls back to host numpy, including the all-NaN
    warning cases)."""
    if not isinstance(x, DeviceArray) or _dtype_code(x.dtype) is None:
        return None
    mask = x.isnan()
    if not isinstance(mask, BoolDeviceArray):
        return None
    n_nan = int(mask.sum())
    n_valid = x.size - n_nan
    if kind in ("max", "min"):
        if n_valid == 0:
            return None  # numpy warns and returns nan
        if n_nan == 0:
            return x.max() if kind == "max" else x.min()
        repl = -_np.inf if kind == "max" else _np.inf
        cleaned = where_device(mask, repl, x)
        if cleaned is NotImplemented:
            return None
        return cleaned.max() if kind == "max" else cleaned.min()
    cleaned = x if n_nan == 0 else where_device(mask, 0.0, x)
    if cleaned is NotImplemented:
        return None
    total = float(cleaned.sum())
    if kind == "sum":
        return x.dtype.type(total)
    if n_valid == 0:
        return None
    mu = total / n_valid
    if kind == "mean":
        return x.dtype.type(mu)
    centered = x._binary("subtract", mu)
    if centered is NotImplemented:
        return None
    cc = centered if n_nan == 0 else where_device(mask, 0.0, centered)
    if cc is NotImplemented:
        return None
    if n_valid - ddof <= 0:
        return None
    var = float(cc.square_sum()) / (n_valid - ddof)
    if kind == "var":
        return x.dtype.type(var)
    return x.dtype.type(var ** 0.5)  # std


def cov_device(m):
    """np.cov for a 2-D DeviceArray (rowvar=True, ddof=1 defaults),
    composed from device ops: row means -> broadcast center -> X @ X.T
    via the device transpose + GEMM. Returns None when not routable."""
    if not isinstance(m, DeviceArray) or len(m.shape) != 2:
        return None
    if _dtype_code(m.dtype) is None:
        return None
    nvar, nobs = m.shape
    if nobs < 2:
        return None
    mu = m.mean(axis=1, keepdims=True)
    if not isinstance(mu, DeviceArray):
        return None  # small-outer host fallback in the axis reducer
    xc = m._binary_bcast("subtract", mu)
    if xc is NotImplemented:
        return None
    prod = matmul(xc, xc._device_transposed(), _force=True)
    if prod is NotImplemented or not isinstance(prod, DeviceArray):
        return None
    r = prod._binary("divide", float(nobs - 1))
    return None if r is NotImplemented else r


def corrcoef_device(m):
    """np.corrcoef via cov_device: diagonal extracted with ONE strided
    download, normalization and the final [-1, 1] clip on device."""
    c = cov_device(m)
    if c is None:
        return None
    n = c.shape[0]
    esz = c.dtype.itemsize
    raw = backend().download_strided(
        c._dev_handle(), 0, (n + 1) * esz, esz, n
    )
    d = _np.sqrt(_np.frombuffer(raw, dtype=c.dtype))
    drow = DeviceArray(backend().upload(_np.ascontiguousarray(d)),
                       (n,), c.dtype)
    dcol = DeviceArray(backend().upload(_np.ascontiguousarray(d)),
                       (n, 1), c.dtype)
    r = c._binary_bcast("divide", dcol)
    if r is NotImplemented:
        return None
    r = r._binary_bcast("divide", drow)
    if r is NotImplemented:
        return None
    return r.clip(-1.0, 1.0)


def quantile_list_device(x, qs):
    """np.quantile/percentile with an ARRAY of quantiles (axis=None):
    ONE device sort, then a 16-byte download_slice per interpolation
    endpoint — exact numpy-linear values at any number of q's."""
    if not isinstance(x, DeviceArray) or _dtype_code(x.dtype) is None:
        return None
    if x.size < 1:
        return None
    if len(x.shape) == 1:
        flat = x
    else:
        flat = x.ravel()  # device copy with its own handle
        if not isinstance(flat, DeviceArray):
            return None
    srt = flat._device_sorted(False)
    esz = x.dtype.itemsize
    n = x.size
    # NaNs sort last: any NaN makes every quantile NaN (numpy parity)
    top = _np.frombuffer(
        backend().download_slice(srt._dev_handle(), (n - 1) * esz, esz),
        dtype=x.dtype,
    )[0]
    out = _np.empty(len(qs), dtype=x.dtype)
    if _np.isnan(top):
        out[:] = _np.nan
        return out
    for i, q in enumerate(qs):
        pos = float(q) * (n - 1)
        k0 = int(_np.floor(pos))
        k1 = min(k0 + 1, n - 1)
        frac = pos - k0
        raw = backend().download_slice(
            srt._dev_handle(), k0 * esz, (k1 - k0 + 1) * esz
        )
        vals = _np.frombuffer(raw, dtype=x.dtype)
        v0 = vals[0]
        v1 = vals[-1]
        out[i] = v0 if frac == 0 else v0 + (v1 - v0) * x.dtype.type(frac)
    return out


def nanquantile_device(x, qs):
    """Flat NaN-ignoring quantiles: one device sort (NaNs sort last),
    quantile positions over the leading non-NaN run, two-element
    download_slice per endpoint. Returns an array aligned with qs, or
    None when not routable (incl. the all-NaN warning case)."""
    if not isinstance(x, DeviceArray) or _dtype_code(x.dtype) is None:
        return None
    if x.size < 1:
        return None
    mask = x.isnan()
    if not isinstance(mask, BoolDeviceArray):
        return None
    n_valid = x.size - int(mask.sum())
    if n_valid == 0:
        return None  # numpy warns and returns nan
    if len(x.shape) == 1:
        flat = x
    else:
        flat = x.ravel()
        if not isinstance(flat, DeviceArray):
            return None
    srt = flat._device_sorted(False)
    esz = x.dtype.itemsize
    out = _np.empty(len(qs), dtype=x.dtype)
    for i, q in enumerate(qs):
        if not 0.0 <= float(q) <= 1.0:
            return None
        pos = float(q) * (n_valid - 1)
        k0 = int(_np.floor(pos))
        k1 = min(k0 + 1, n_valid - 1)
        frac = pos - k0
        raw = backend().download_slice(
            srt._dev_handle(), k0 * esz, (k1 - k0 + 1) * esz
        )
        vals = _np.frombuffer(raw, dtype=x.dtype)
        v0, v1 = vals[0], vals[-1]
        out[i] = v0 if frac == 0 else v0 + (v1 - v0) * x.dtype.type(frac)
    return out


def quantile_cols_device(x, q):
    """Per-column quantile (axis=0): device transpose, then the row
    path."""
    if not isinstance(x, DeviceArray) or len(x.shape) != 2:
        return None
    if _dtype_code(x.dtype) is None or not 0.0 <= q <= 1.0:
        return None
    return quantile_rows_device(x._device_transposed(), q)


def quantile_device(x, q):
    """numpy-linear-interpolation quantile of a DeviceArray (axis=None).
    Returns None when not routable (caller falls back to host)."""
    if not isinstance(x, DeviceArray) or isinstance(x, BoolDeviceArray):
        return None
    if _dtype_code(x.dtype) is None or x.size == 0:
        return None
    try:
        q = float(q)
    except (TypeError, ValueError):
        return None
    if not 0.0 <= q <= 1.0:
        return None
    h = (x.size - 1) * q
    k0 = int(_np.floor(h))
    k1 = int(_np.ceil(h))
    v0, nan_c = _order_stat_device(x, k0)
    if nan_c:
        return x.dtype.type("nan")
    if v0 != v0:  # non-convergence sentinel without NaNs: fall back
        return None
    if k1 == k0:
        return x.dtype.type(v0)
    v1, nan_c = _order_stat_device(x, k1)
    if nan_c:
        return _np.float64("nan")
    if v1 != v1:
        return None
    # numpy returns the input's dtype for median/quantile of floats
    return x.dtype.type(v0 + (h - k0) * (v1 - v0))


def where_device(mask, a, b):
    """np.where(mask, a, b) on device; NotImplemented if not routable."""
    if not isinstance(mask, BoolDeviceArray):
        return NotImplemented

    def classify(x):
        if isinstance(x, DeviceArray) and not isinstance(x, BoolDeviceArray):
            if x.shape == mask.shape and _dtype_code(x.dtype) is not None:
                return x, None
            return None, None  # unroutable array
        if isinstance(x, (bool, _np.bool_)):
            return None, None
        if isinstance(x, (int, float)):
            return None, float(x)
        return None, None

    da, sa = classify(a)
    db, sb = classify(b)
    if (da is None and sa is None) or (db is None and sb is None):
        return NotImplemented
    if da is not None and db is not None and da.dtype != db.dtype:
        return NotImplemented
    out_dtype = da.dtype if da is not None else (
        db.dtype if db is not None else _np.dtype(_np.float64)
    )
    code = _dtype_code(out_dtype)
    h = backend().where(
        mask._dev_handle(), code, mask.size,
        da._dev_handle() if da is not None else 0, sa or 0.0,
        db._dev_handle() if db is not None else 0, sb or 0.0,
    )
    return DeviceArray(h, mask.shape, out_dtype)


def _asarray(x):
    return x.materialize() if isinstance(x, DeviceArray) else _np.asarray(x)


def _asarray_or_scalar(x):
    return x.materialize() if isinstance(x, DeviceArray) else x


def _to_device(x) -> "DeviceArray | None":
    """Upload a host ndarray (f32/f64, C-contiguous) to the device."""
    if isinstance(x, DeviceArray):
        return x
    arr = _np.asarray(x)
    if _dtype_code(arr.dtype) is None:
        return None
    arr = _np.ascontiguousarray(arr)
    return DeviceArray(backend().upload(arr), arr.shape, arr.dtype)


# ---------------------------------------------------------------------------
# module-level compute entry points (used by the numpy patches and tests)
# ---------------------------------------------------------------------------
def rand(*shape, seed=None):
    """Uniform [0,1) float64 of the given shape, generated on-device
    (Philox4x32-10)."""
    _ensure_ready()
    n = 1
    for s in shape:
        n *= int(s)
    if seed is None:
        seed = int(_np.random.randint(0, 2**63 - 1, dtype=_np.int64))
    h = backend().rand(n, _F64, int(seed))
    return DeviceArray(h, shape if shape else (), _np.float64)


def normal_device(loc: float, scale: float, *shape, seed=None) -> "DeviceArray":
    """N(loc, scale^2) float64 on-device: Philox + Box-Muller with the
    affine transform fused into the generating kernel."""
    _ensure_ready()
    n = 1
    for sh in shape:
        n *= int(sh)
    if seed is None:
        seed = int(_np.random.randint(0, 2**63 - 1, dtype=_np.int64))
    h = backend().randn(n, int(seed), float(loc), float(scale))
    return DeviceArray(h, shape if shape else (), _np.float64)


def uniform_device(low: float, high: float, size) -> "DeviceArray":
    """np.random.uniform semantics on-device: rand * (high-low) + low."""
    shape = (size,) if isinstance(size, int) else tuple(size)
    x = rand(*shape)
    if low == 0.0 and high == 1.0:
        return x
    scaled = x._binary("multiply", float(high - low))
    return scaled._binary("add", float(low))


def square(x):
    _ensure_ready()
    d = _to_device(x)
    if d is None:
        return _np.square(_asarray(x))
    return d._unary("square")


def sum_(x):
    _ensure_ready()
    d = _to_device(x)
    if d is None:
        return _np.sum(_asarray(x))
    return d.sum()


def square_sum(x):
    _ensure_ready()
    d = _to_device(x)
    if d is None:
        return _np.sum(_np.square(_asarray(x)))
    return d.square_sum()


def matmul(a, b, _force=False):
    """Row-major 2D matmul on the MFMA matrix cores (f32: 32x32x2 f32 MFMA,
    f64: 16x16x4 f64 MFMA); equal-batch 3D stacks run per-batch GEMMs
    enqueued back-to-back. Returns NotImplemented when the shape/dtype is
    not routable (caller falls back)."""
    _ensure_ready()
    a_shape = a.shape if hasattr(a, "shape") else _np.asarray(a).shape
    b_shape = b.shape if hasattr(b, "shape") else _np.asarray(b).shape
    if (
        len(a_shape) == 3
        and len(b_shape) == 3
        and a_shape[0] == b_shape[0]
        and a_shape[2] == b_shape[1]
    ):
        return _matmul_batched(a, b, _force)
    if (
        len(a_shape) == 1
        and b_shape == a_shape
        and a_shape[0] >= MIN_ELEMS
    ):
        da, db = _to_device(a), _to_device(b)
        if da is not None and db is not None and da.dtype == db.dtype:
            prod = da._binary("multiply", db)
            if prod is not NotImplemented:
                return prod.sum()  # 1-D dot: scalar on device
    if (
        len(a_shape) == 2
        and len(b_shape) == 1
        and a_shape[1] == b_shape[0]
    ):
        # matvec: x is bit-identical to a (k, 1) column matrix
        m, k = a_shape
        if _force or 2.0 * m * k >= MIN_MATMUL_FLOPS:
            da, db = _to_device(a), _to_device(b)
            if da is not None and db is not None and da.dtype == db.dtype:
                code = _dtype_code(da.dtype)
                if code is not None:
                    hc = backend().gemm(
                        da._dev_handle(), db._dev_handle(), m, 1, k, code
                    )
                    return DeviceArray(hc, (m,), da.dtype)
        return NotImplemented
    if (
        len(a_shape) == 1
        and len(b_shape) == 2
        and a_shape[0] == b_shape[0]
    ):
        # vecmat: x is a (1, k) row matrix
        k, n = b_shape
        if _force or 2.0 * n * k >= MIN_MATMUL_FLOPS:
            da, db = _to_device(a), _to_device(b)
            if da is not None and db is not None and da.dtype == db.dtype:
                code = _dtype_code(da.dtype)
                if code is not None:
                    hc = backend().gemm(
                        da._dev_handle(), db._dev_handle(), 1, n, k, code
                    )
                    return DeviceArray(hc, (n,), da.dtype)
        return NotImplemented
    if len(a_shape) != 2 or len(b_shape) != 2 or a_shape[1] != b_shape[0]:
        return NotImplemented
    m, k = a_shape
    n = b_shape[1]
    if not _force and 2.0 * m * n * k < MIN_MATMUL_FLOPS:
        return NotImplemented
    da = _to_device(a)
    db = _to_device(b)
    if da is None or db is None or da.dtype != db.dtype:
        return NotImplemented
    code = _dtype_code(da.dtype)
    if code is None:  # e.g. boolean matmul: numpy semantics on host
        return NotImplemented
    hc = backend().gemm(da._dev_handle(), db._dev_handle(), m, n, k, code)
    return DeviceArray(hc, (m, n), da.dtype)


def _matmul_batched(a, b, _force=False):
    """[batch][m][k] @ [batch][k][n] via per-batch device GEMMs."""
    a_shape = a.shape if hasattr(a, "shape") else _np.asarray(a).shape
    b_shape = b.shape if hasattr(b, "shape") else _np.asarray(b).shape
    batch, m, k = a_shape
    n = b_shape[2]
    if not _force and 2.0 * batch * m * n * k < MIN_MATMUL_FLOPS:
        return NotImplemented
    da = _to_device(a)
    db = _to_device(b)
    if da is None or db is None or da.dtype != db.dtype:
        return NotImplemented
    code = _dtype_code(da.dtype)
    if code is None:
        return NotImplemented
    hc = backend().gemm_batched(
        da._dev_handle(), db._dev_handle(), batch, m, n, k, code
    )
    return DeviceArray(hc, (batch, m, n), da.dtype)


# ---------------------------------------------------------------------------
# numpy patching
# ---------------------------------------------------------------------------
_installed = {"done": False}


def install(numpy_module, mode: str = "auto") -> None:
    """Patch numpy's hot entry points to route large work to the GPU.
    mode="require" raises if the GPU/extension is unusable."""
    if _installed["done"]:
        return
    if not available():
        if mode == "require":
            raise RuntimeError("APP_HIP_NUMPY=require but no AMD GPU is visible")
        return

    np = numpy_module
    orig_rand = np.random.rand
    orig_random = np.random.random
    orig_random_sample = np.random.random_sample
    orig_uniform = np.random.uniform
    orig_matmul = np.matmul
    orig_dot = np.dot
    orig_square = np.square
    orig_sum = np.sum

    def patched_rand(*shape):
        n = 1
        for s in shape:
            n *= int(s)
        if n >= MIN_ELEMS:
            try:
                return rand(*shape)
            except Exception:
                if mode == "require":
                    raise
        return orig_rand(*shape)

    def _size_elems(size):
        if size is None:
            return 1
        if isinstance(size, int):
            return size
        n = 1
        for s in size:
            n *= int(s)
        return n

    def patched_random(size=None):
        if size is not None and _size_elems(size) >= MIN_ELEMS:
            try:
                shape = (size,) if isinstance(size, int) else tuple(size)
                return rand(*shape)
            except Exception:
                if mode == "require":
                    raise
        return orig_random(size)

    def patched_uniform(low=0.0, high=1.0, size=None):
        if (
            size is not None
            and _size_elems(size) >= MIN_ELEMS
            and isinstance(low, (int, float))
            and isinstance(high, (int, float))
        ):
            try:
                return uniform_device(float(low), float(high), size)
            except Exception:
                if mode == "require":
                    raise
        return orig_uniform(low, high, size)

    def patched_matmul(a, b, *args, **kwargs):
        if not args and not kwargs:
            try:
                r = matmul(a, b)
                if r is not NotImplemented:
                    return r
            except Exception:
                if mode == "require":
                    raise
        return orig_matmul(_asarray(a), _asarray(b), *args, **kwargs)

    def patched_dot(a, b, *args, **kwargs):
        if not args and not kwargs:
            try:
                r = matmul(a, b)
                if r is not NotImplemented:
                    return r
            except Exception:
                if mode == "require":
                    raise
        return orig_dot(_asarray(a), _asarray(b), *args, **kwargs)

    def _make_patched_unary(opname, orig):
        """Module-level unary routing (np.exp/log/sqrt/... on large host
        arrays): upload at PCIe rate + one kernel beats host
        transcendental loops ~20-50x at 1e8 elements; DeviceArray inputs
        stay resident. Same passthrough pattern as patched_square."""

        def patched(x, *args, **kwargs):
            if not args and not kwargs:
                if isinstance(x, DeviceArray) and not isinstance(
                    x, BoolDeviceArray
                ):
                    try:
                        return x._unary(opname)
                    except Exception:
                        if mode == "require":
                            raise
                arr = x if isinstance(x, _np.ndarray) else None
                if (
                    arr is not None
                    and arr.size >= MIN_ELEMS
                    and _dtype_code(arr.dtype) is not None
                    and arr.flags.c_contiguous
                ):
                    try:
                        d = _to_device(arr)
                        if d is not None:
                            return d._unary(opname)
                    except Exception:
                        if mode == "require":
                            raise
            return orig(_asarray(x), *args, **kwargs)

        return patched

    patched_square = _make_patched_unary("square", orig_square)

    def patched_sum(x, *args, **kwargs):
        if isinstance(x, DeviceArray) and not args and set(kwargs) <= {
            "axis", "keepdims"
        }:
            # the method handles axis/keepdims on device and falls back
            # itself; going through orig_sum would materialize
            return x.sum(**kwargs)
        if not args and (not kwargs or set(kwargs) <= {"axis"}) and kwargs.get("axis") is None:
            arr = x if isinstance(x, _np.ndarray) else None
            if (
                arr is not None
                and arr.size >= MIN_ELEMS
                and _dtype_code(arr.dtype) is not None
                and arr.flags.c_contiguous
            ):
                try:
                    return sum_(arr)
                except Exception:
                    if mode == "require":
                        raise
        return orig_sum(_asarray(x), *args, **kwargs)

    orig_randn = np.random.randn
    orig_standard_normal = np.random.standard_normal
    orig_normal = np.random.normal

    def patched_randn(*shape):
        n = 1
        for sh in shape:
            n *= int(sh)
        if n >= MIN_ELEMS:
            try:
                return normal_device(0.0, 1.0, *shape)
            except Exception:
                if mode == "require":
                    raise
        return orig_randn(*shape)

    def patched_standard_normal(size=None):
        if size is not None and _size_elems(size) >= MIN_ELEMS:
            try:
                shape = (size,) if isinstance(size, int) else tuple(size)
                return normal_device(0.0, 1.0, *shape)
            except Exception:
                if mode == "require":
                    raise
        return orig_standard_normal(size)

    def patched_normal(loc=0.0, scale=1.0, size=None):
        if (
            size is not None
            and _size_elems(size) >= MIN_ELEMS
            and isinstance(loc, (int, float))
            and isinstance(scale, (int, float))
        ):
            try:
                shape = (size,) if isinstance(size, int) else tuple(size)
                return normal_device(float(loc), float(scale), *shape)
            except Exception:
                if mode == "require":
                    raise
        return orig_normal(loc, scale, size)

    np.random.randn = patched_randn
    np.random.standard_normal = patched_standard_normal
    np.random.normal = patched_normal
    np.random.rand = patched_rand
    np.random.random = patched_random
    np.random.random_sample = patched_random
    np.random.uniform = patched_uniform
    np.matmul = patched_matmul
    np.dot = patched_dot
    np.square = patched_square
    np.sum = patched_sum
    orig_sort = np.sort
    orig_argsort = np.argsort
    orig_median = np.median

    def _promote_big(x):
        """Upload a large host ndarray (PCIe upload + device op beats a
        host O(n log n) pass by 10-100x at >=MIN_ELEMS)."""
        if (
            isinstance(x, _np.ndarray)
            and x.size >= MIN_ELEMS
            and _dtype_code(x.dtype) is not None
            and x.flags.c_contiguous
        ):
            return _to_device(x)
        return None

    def patched_sort(a, axis=-1, kind=None, order=None, **kw):
        # numpy 2.x adds a keyword-only `stable`; stable=True is what the
        # radix sort IS, so it routes too
        routable_kw = not kw or (set(kw) == {"stable"})
        if not isinstance(a, DeviceArray) and order is None and routable_kw:
            try:
                d = _promote_big(a)
                if d is not None and d._sort_routable(axis, kind, order, {}):
                    return d._device_sorted(False, axis=d._sort_axis01(axis))
            except Exception:
                if mode == "require":
                    raise
        return orig_sort(a, axis=axis, kind=kind, order=order, **kw)

    def patched_argsort(a, axis=-1, kind=None, order=None, **kw):
        routable_kw = not kw or (set(kw) == {"stable"})
        if not isinstance(a, DeviceArray) and order is None and routable_kw:
            try:
                d = _promote_big(a)
                if d is not None and d._sort_routable(axis, kind, order, {}):
                    _, idx = d._device_sorted(
                        True, axis=d._sort_axis01(axis)
                    )
                    return idx
            except Exception:
                if mode == "require":
                    raise
        return orig_argsort(a, axis=axis, kind=kind, order=order, **kw)

    def patched_median(a, axis=None, **kwargs):
        if not isinstance(a, DeviceArray) and not kwargs:
            try:
                d = _promote_big(a)
                if d is not None:
                    if axis is None:
                        r = quantile_device(d, 0.5)
                    elif axis in (1, -1) and len(d.shape) == 2:
                        r = quantile_rows_device(d, 0.5)
                    elif axis == 0 and len(d.shape) == 2:
                        r = quantile_cols_device(d, 0.5)
                    else:
                        r = None
                    if r is not None:
                        return r
            except Exception:
                if mode == "require":
                    raise
        return orig_median(a, axis=axis, **kwargs)

    np.sort = patched_sort
    np.argsort = patched_argsort
    np.median = patched_median

    def _make_patched_scalar_reduce(meth, orig):
        """np.mean/std/var/max/min on large HOST arrays: upload + one
        fused device reduce beats the host pass 4-40x; returns the same
        scalar. DeviceArray inputs go through the method (device axis
        handling + fallback)."""

        def patched(x, *args, **kwargs):
            if isinstance(x, DeviceArray) and not args:
                allowed = {"axis", "keepdims"} | (
                    {"ddof"} if meth in ("std", "var") else set()
                )
                if set(kwargs) <= allowed:
                    return getattr(x, meth)(**kwargs)
            if not args and not kwargs:
                try:
                    d = _promote_big(x)
                    if d is not None:
                        return getattr(d, meth)()
                except Exception:
                    if mode == "require":
                        raise
            return orig(_asarray(x), *args, **kwargs)

        return patched

    for _meth, _names in (
        ("mean", ("mean",)), ("std", ("std",)), ("var", ("var",)),
        ("max", ("max", "amax")), ("min", ("min", "amin")),
    ):
        for _n in _names:
            setattr(np, _n, _make_patched_scalar_reduce(
                _meth, getattr(np, _n)))

    # the rest of the hot unary surface, same pattern (np.abs is an
    # alias of np.absolute; both get the patch)
    for _uname, _npname in (
        ("sqrt", "sqrt"), ("exp", "exp"), ("log", "log"), ("sin", "sin"),
        ("cos", "cos"), ("tanh", "tanh"), ("absolute", "absolute"),
    ):
        _orig = getattr(np, _npname)
        _patched = _make_patched_unary(_uname, _orig)
        setattr(np, _npname, _patched)
        if _npname == "absolute":
            np.abs = _patched
    _installed["done"] = True


# ---------------------------------------------------------------------------
# __array_function__ dispatch stages. Each stage handles a family of
# numpy functions and returns _AF_PASS to fall through to the generic
# materialize-and-delegate fallback. Kept as plain module functions so
# each family reads (and diffs) independently.
# ---------------------------------------------------------------------------
_AF_PASS = object()
# identity snapshots taken at import time: install() REPLACES
# numpy.sort/argsort/median module attributes, but the NEP-18 protocol
# dispatches with the ORIGINAL function object (the wrappers call the
# saved originals), so handlers must recognize both
_NP_SORT0 = _np.sort
_NP_ARGSORT0 = _np.argsort
_NP_MEDIAN0 = _np.median
_SORT_FUNCS = (_NP_SORT0, _np.sort)
_ARGSORT_FUNCS = (_NP_ARGSORT0, _np.argsort)
_MEDIAN_FUNCS = (_NP_MEDIAN0, _np.median)
_SUM_FUNCS = (_np.sum,)
_MATMUL_FUNCS = (_np.matmul, _np.dot)
_SQUARE_FUNCS = (_np.square,)
def _af_linalg(func, args, kwargs):
    if func in _SUM_FUNCS and len(args) == 1 and isinstance(args[0], DeviceArray):
        if set(kwargs) <= {"axis", "keepdims"}:
            return args[0].sum(**kwargs)
    if func is _np.mean and len(args) == 1 and isinstance(args[0], DeviceArray):
        if set(kwargs) <= {"axis", "keepdims"}:
            return args[0].mean(**kwargs)
    if func is _np.trace and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        a = args[0]
        if (
            len(a.shape) == 2
            and _dtype_code(a.dtype) is not None
            and a._host is None
        ):
            n = min(a.shape)
            esz = a.dtype.itemsize
            raw = backend().download_strided(
                a._dev_handle(), 0, (a.shape[1] + 1) * esz, esz, n
            )
            return a.dtype.type(
                _np.frombuffer(raw, dtype=a.dtype).sum()
            )
        return _np.trace(a.materialize())
    if func is _np.outer and len(args) == 2 and not kwargs:
        x, y = args
        if (
            isinstance(x, DeviceArray)
            and isinstance(y, DeviceArray)
            and len(x.shape) == 1
            and len(y.shape) == 1
            and x.dtype == y.dtype
            and _dtype_code(x.dtype) is not None
        ):
            h = backend().gemm(
                x._dev_handle(), y._dev_handle(), x.size, y.size, 1,
                _dtype_code(x.dtype),
            )
            return DeviceArray(h, (x.size, y.size), x.dtype)
        host = [
            v.materialize() if isinstance(v, DeviceArray) else v
            for v in args
        ]
        return _np.outer(*host)
    if func is _np.einsum and len(args) >= 2 and isinstance(
        args[0], str
    ) and not kwargs:
        # the common contractions, mapped onto existing device ops;
        # anything else materializes below
        sub = args[0].replace(" ", "")
        ops_ = args[1:]
        try:
            if sub in ("ij,jk->ik", "ij,jk") and len(ops_) == 2:
                r = matmul(ops_[0], ops_[1], _force=True)
                if r is not NotImplemented:
                    return r
            elif sub in ("bij,bjk->bik", "bij,bjk") and len(ops_) == 2:
                r = matmul(ops_[0], ops_[1], _force=True)
                if r is not NotImplemented:
                    return r
            elif sub in ("i,i->", "i,i") and len(ops_) == 2:
                r = matmul(ops_[0], ops_[1], _force=True)
                if r is not NotImplemented:
                    return r
            elif sub in ("ij->ji",) and len(ops_) == 1 and isinstance(
                ops_[0], DeviceArray
            ):
                return ops_[0].transpose()
            elif sub in ("ij->", "i->") and len(ops_) == 1 and isinstance(
                ops_[0], DeviceArray
            ):
                return ops_[0].sum()
            elif sub == "ii" and len(ops_) == 1:
                return DeviceArray.__array_function__(
                    ops_[0], _np.trace, (DeviceArray,), (ops_[0],), {}
                )
            elif sub in ("i,j->ij", "i,j") and len(ops_) == 2:
                return DeviceArray.__array_function__(
                    ops_[0], _np.outer, (DeviceArray,), tuple(ops_), {}
                )
        except Exception:
            pass
        host = [
            v.materialize() if isinstance(v, DeviceArray) else v
            for v in ops_
        ]
        return _np.einsum(sub, *host)
    if func in _MATMUL_FUNCS and len(args) == 2 and not kwargs:
        r = matmul(*args, _force=True)
        if r is not NotImplemented:
            return r
    if func in _SQUARE_FUNCS and len(args) == 1 and isinstance(args[0], DeviceArray):
        return args[0]._unary("square")
    return _AF_PASS

def _af_order_stats(func, args, kwargs):
    if func in _MEDIAN_FUNCS and len(args) == 1 and not kwargs:
        r = quantile_device(args[0], 0.5)
        if r is not None:
            return r
    if func in _MEDIAN_FUNCS and len(args) == 1 and set(kwargs) == {"axis"}:
        if kwargs["axis"] in (1, -1):
            r = quantile_rows_device(args[0], 0.5)
            if r is not None:
                return r
        elif kwargs["axis"] == 0:
            r = quantile_cols_device(args[0], 0.5)
            if r is not None:
                return r
        a0 = args[0]
        if isinstance(a0, DeviceArray):
            return _np.median(a0.materialize(), **kwargs)
    if func in (_np.quantile, _np.percentile) and len(args) == 2 and not kwargs:
        qv = args[1]
        if isinstance(qv, (int, float)):
            q = qv / 100.0 if func is _np.percentile else float(qv)
            r = quantile_device(args[0], q)
            if r is not None:
                return r
        elif isinstance(qv, (list, tuple, _np.ndarray)):
            qarr = _np.asarray(qv, dtype=_np.float64).reshape(-1)
            if qarr.size and _np.all((qarr >= 0) & (qarr <= 100)):
                qs = qarr / 100.0 if func is _np.percentile else qarr
                if _np.all(qs <= 1.0):
                    r = quantile_list_device(args[0], qs)
                    if r is not None:
                        return r.reshape(_np.asarray(qv).shape)
    if func in (_np.quantile, _np.percentile) and len(args) == 2 and set(
        kwargs
    ) == {"axis"}:
        qv = args[1]
        if isinstance(qv, (int, float)) and kwargs["axis"] in (0, 1, -1):
            q = qv / 100.0 if func is _np.percentile else float(qv)
            if kwargs["axis"] == 0:
                r = quantile_cols_device(args[0], q)
            else:
                r = quantile_rows_device(args[0], q)
            if r is not None:
                return r
        elif isinstance(qv, (list, tuple, _np.ndarray)) and kwargs[
            "axis"
        ] in (0, 1, -1):
            qarr = _np.asarray(qv, dtype=_np.float64).reshape(-1)
            if qarr.size and _np.all((qarr >= 0) & (qarr <= 100)):
                qs = qarr / 100.0 if func is _np.percentile else qarr
                if _np.all(qs <= 1.0):
                    a0 = args[0]
                    if kwargs["axis"] == 0 and isinstance(
                        a0, DeviceArray
                    ) and len(a0.shape) == 2:
                        a0 = a0._device_transposed()
                    r = quantile_rows_device_multi(a0, qs)
                    if r is not None:
                        return r
        a0 = args[0]
        if isinstance(a0, DeviceArray):
            return func(a0.materialize(), qv, **kwargs)
    if func is _np.where and len(args) == 3 and not kwargs:
        r = where_device(*args)
        if r is not NotImplemented:
            return r
    if func is _np.count_nonzero and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ):
        if not kwargs:
            if isinstance(args[0], BoolDeviceArray):
                return args[0].sum()
            return args[0].count_nonzero()
    return _AF_PASS

def _af_array_ops(func, args, kwargs):
    if func is _np.cumsum and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ):
        if set(kwargs) <= {"axis"}:
            return args[0].cumsum(**kwargs)
    if func in (_np.any, _np.all) and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ):
        if not kwargs:
            meth = "any" if func is _np.any else "all"
            return getattr(args[0], meth)()
    if func is _np.clip and len(args) >= 1 and isinstance(args[0], DeviceArray):
        if len(args) <= 3 and not kwargs:
            return args[0].clip(*args[1:])
    if func in (_np.argmax, _np.argmin) and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ):
        if set(kwargs) <= {"axis"}:
            meth = "argmax" if func is _np.argmax else "argmin"
            return getattr(args[0], meth)(**kwargs)
    if func in (_np.partition, _np.argpartition) and len(args) == 2 \
            and isinstance(args[0], DeviceArray):
        a, kth = args
        axis = kwargs.get("axis", -1)
        order = kwargs.get("order")
        extra = {
            k: v for k, v in kwargs.items()
            if k not in ("axis", "kind", "order")
        }
        if a._sort_routable(axis, None, order, extra):
            ax01 = a._sort_axis01(axis)
            if func is _np.partition:
                return a._device_sorted(False, axis=ax01)
            _, idx = a._device_sorted(True, axis=ax01)
            return idx
        return func(a.materialize(), kth, **kwargs)
    if func is _np.unique and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        a = args[0]
        if a._sort_routable(None, None, None, {}):
            # numpy's unique IS sort + adjacent-compare
            # (lib/arraysetops); NaNs sort last and collapse to one
            # (equal_nan=True is the numpy>=1.21 default)
            host = a._device_sorted(False).materialize()
            if host.size == 0:
                return host
            keep = _np.empty(host.size, dtype=bool)
            keep[0] = True
            _np.not_equal(host[1:], host[:-1], out=keep[1:])
            n_nan = int(_np.isnan(host[-1:])[0] and
                        _np.isnan(host).sum())
            if n_nan > 1:
                keep[host.size - n_nan + 1:] = False
                keep[host.size - n_nan] = True
            return host[keep]
        return _np.unique(a.materialize())
    if func in _SORT_FUNCS + _ARGSORT_FUNCS and len(args) == 1 \
            and isinstance(args[0], DeviceArray):
        a = args[0]
        axis = kwargs.get("axis", -1)
        kind = kwargs.get("kind")
        order = kwargs.get("order")
        extra = {
            k: v for k, v in kwargs.items()
            if k not in ("axis", "kind", "order")
        }
        if a._sort_routable(axis, kind, order, extra):
            ax01 = a._sort_axis01(axis)
            if func in _SORT_FUNCS:
                return a._device_sorted(False, axis=ax01)
            _, idx = a._device_sorted(True, axis=ax01)
            return idx
        host = a.materialize()
        return func(host, **kwargs)
    if func is _np.searchsorted and len(args) in (2, 3) and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"side"}:
        a = args[0]
        v = args[1]
        side = args[2] if len(args) == 3 else kwargs.get("side", "left")
        code = _dtype_code(a.dtype)
        if (
            code is not None
            and len(a.shape) == 1
            and side in ("left", "right")
        ):
            scalar_q = _np.isscalar(v) or (
                isinstance(v, _np.ndarray) and v.ndim == 0
            )
            if isinstance(v, DeviceArray):
                if v.dtype == a.dtype and len(v.shape) >= 1:
                    hv, m, vshape = v._dev_handle(), v.size, v.shape
                else:
                    hv = None
            elif scalar_q or isinstance(v, (list, tuple, _np.ndarray)):
                host_v = _np.ascontiguousarray(
                    _np.asarray(v, dtype=a.dtype).reshape(-1)
                )
                if host_v.size >= 1:
                    # wrapped so the staging buffer is freed on GC
                    _tmp_q = DeviceArray(
                        backend().upload(host_v), host_v.shape, a.dtype
                    )
                    hv, m, vshape = _tmp_q._handle, \
                        host_v.size, _np.asarray(v).shape
                else:
                    hv = None
            else:
                hv = None
            if hv is not None:
                h = backend().searchsorted(
                    a._dev_handle(), a.shape[0], hv, m, code,
                    1 if side == "right" else 0,
                )
                res = DeviceArray(h, (m,), _np.int64)
                if scalar_q:
                    out = _np.intp(res.materialize()[0])
                    return out
                if vshape != (m,):
                    return res.materialize().reshape(vshape)
                return res
        host = a.materialize()
        return _np.searchsorted(host, *args[1:], **kwargs)
    if func is _np.digitize and len(args) == 2 and set(kwargs) <= {
        "right"
    }:
        # numpy: digitize(x, bins) == searchsorted(bins, x, side=...)
        # for monotonically increasing bins
        x, bins = args
        right = bool(kwargs.get("right", False))
        if isinstance(x, DeviceArray) and len(x.shape) == 1:
            b = _np.asarray(
                bins.materialize() if isinstance(bins, DeviceArray) else bins
            )
            if (
                b.ndim == 1
                and b.size >= 1
                and _np.all(b[1:] >= b[:-1])  # increasing bins only
                and _dtype_code(x.dtype) is not None
            ):
                db = _to_device(_np.ascontiguousarray(b.astype(x.dtype)))
                if db is not None:
                    h = backend().searchsorted(
                        db._dev_handle(), b.size, x._dev_handle(), x.size,
                        _dtype_code(x.dtype), 0 if right else 1,
                    )
                    return DeviceArray(h, x.shape, _np.int64)
        host_x = x.materialize() if isinstance(x, DeviceArray) else x
        host_b = bins.materialize() if isinstance(bins, DeviceArray) else bins
        return _np.digitize(host_x, host_b, **kwargs)
    if func is _np.diff and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"axis", "n"} and kwargs.get("n", 1) == 1:
        a = args[0]
        axis = kwargs.get("axis", -1)
        code = _dtype_code(a.dtype)
        if code is not None and len(a.shape) in (1, 2):
            nd = len(a.shape)
            ax = a._norm_axis(axis, nd)
            if nd == 1 and ax == 0 and a.shape[0] >= 2:
                h = backend().diff(a._dev_handle(), code, 1, a.shape[0])
                return DeviceArray(h, (a.shape[0] - 1,), a.dtype)
            if nd == 2 and ax == 1 and a.shape[1] >= 2:
                rows, cols = a.shape
                h = backend().diff(a._dev_handle(), code, rows, cols)
                return DeviceArray(h, (rows, cols - 1), a.dtype)
            if nd == 2 and ax == 0 and a.shape[0] >= 2:
                t = a._device_transposed()
                rows, cols = t.shape
                h = backend().diff(t._dev_handle(), code, rows, cols)
                return DeviceArray(
                    h, (rows, cols - 1), a.dtype
                )._device_transposed()
        return _np.diff(a.materialize(), **kwargs)
    return _AF_PASS

def _af_cleanup(func, args, kwargs):
    if func is _np.nan_to_num and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"nan", "posinf", "neginf"}:
        a = args[0]
        if _dtype_code(a.dtype) is not None:
            nan_v = float(kwargs.get("nan", 0.0))
            big = float(_np.finfo(a.dtype).max)
            pos_v = float(kwargs.get("posinf", big))
            neg_v = float(kwargs.get("neginf", -big))
            r = a
            m = r.isnan()
            if isinstance(m, BoolDeviceArray) and int(m.sum()):
                r2 = where_device(m, nan_v, r)
                if r2 is not NotImplemented:
                    r = r2
            for bound, repl in ((_np.inf, pos_v), (-_np.inf, neg_v)):
                m = r._compare("equal", bound)
                if m is not None and int(m.sum()):
                    r2 = where_device(m, repl, r)
                    if r2 is not NotImplemented:
                        r = r2
            # numpy always returns a copy; when nothing needed
            # replacing, clone so mutation can't alias the source
            return r if r is not a else DeviceArray(
                a._device_clone(), a.shape, a.dtype
            )
        return _np.nan_to_num(a.materialize(), **kwargs)
    if func in (_np.real, _np.conj, _np.conjugate) and len(args) == 1 \
            and isinstance(args[0], DeviceArray) and not kwargs:
        return args[0]  # real dtypes only ever reach the device
    if func is _np.imag and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        return _np.zeros(args[0].shape, dtype=args[0].dtype)
    if func is _np.ptp and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        a = args[0]
        if _dtype_code(a.dtype) is not None:
            return a.dtype.type(float(a.max()) - float(a.min()))
        return _np.ptp(a.materialize())
    if func is _np.average and len(args) == 1 and set(kwargs) <= {
        "weights"
    }:
        a = args[0]
        w = kwargs.get("weights")
        if (
            isinstance(a, DeviceArray)
            and _dtype_code(a.dtype) is not None
        ):
            if w is None:
                return a.dtype.type(float(a.sum()) / a.size)
            if (
                isinstance(w, DeviceArray)
                and w.shape == a.shape
                and w.dtype == a.dtype
            ):
                num = a._binary("multiply", w)
                if num is not NotImplemented:
                    den = float(w.sum())
                    if den != 0:
                        return a.dtype.type(float(num.sum()) / den)
        host_w = w.materialize() if isinstance(w, DeviceArray) else w
        host_a = a.materialize() if isinstance(a, DeviceArray) else a
        return _np.average(host_a, weights=host_w)
    if func in (_np.isclose, _np.allclose) and len(args) == 2 and set(
        kwargs
    ) <= {"rtol", "atol", "equal_nan"} and not kwargs.get("equal_nan"):
        a, b = args
        rtol = float(kwargs.get("rtol", 1e-05))
        atol = float(kwargs.get("atol", 1e-08))
        if (
            isinstance(a, DeviceArray)
            and isinstance(b, DeviceArray)
            and a.shape == b.shape
            and a.dtype == b.dtype
            and _dtype_code(a.dtype) is not None
        ):
            # |a-b| <= atol + rtol*|b|, NaN comparisons are False
            diff = a._binary("subtract", b)
            if diff is not NotImplemented:
                tol = b._unary("absolute")._binary("multiply", rtol)
                if tol is not NotImplemented:
                    tol = tol._binary("add", atol)
                if tol is not NotImplemented:
                    mask = diff._unary("absolute")._compare(
                        "less_equal", tol
                    )
                    if mask is not None:
                        if func is _np.allclose:
                            return bool(mask.all())
                        return mask
        host = [
            v.materialize() if isinstance(v, DeviceArray) else v
            for v in args
        ]
        return func(*host, **kwargs)
    if func in (_np.nanmedian, _np.nanquantile, _np.nanpercentile) \
            and not kwargs and len(args) in (1, 2) and isinstance(
        args[0], DeviceArray
    ):
        a = args[0]
        qs = None
        scalar_q = True
        if func is _np.nanmedian and len(args) == 1:
            qs = [0.5]
        elif len(args) == 2:
            qv = args[1]
            scale = 100.0 if func is _np.nanpercentile else 1.0
            if isinstance(qv, (int, float)):
                qs = [float(qv) / scale]
            elif isinstance(qv, (list, tuple, _np.ndarray)):
                qs = (_np.asarray(qv, dtype=_np.float64).reshape(-1)
                      / scale).tolist()
                scalar_q = False
        if qs:
            r = nanquantile_device(a, qs)
            if r is not None:
                if scalar_q:
                    return a.dtype.type(r[0])
                return r.reshape(_np.asarray(args[1]).shape)
        return func(a.materialize(), *args[1:], **kwargs)
    if func in (_np.nanargmax, _np.nanargmin) and len(args) == 1 \
            and isinstance(args[0], DeviceArray) and not kwargs:
        a = args[0]
        if _dtype_code(a.dtype) is not None:
            mask = a.isnan()
            if isinstance(mask, BoolDeviceArray):
                n_nan = int(mask.sum())
                if n_nan == 0:
                    return (a.argmax() if func is _np.nanargmax
                            else a.argmin())
                if n_nan < a.size:
                    repl = -_np.inf if func is _np.nanargmax else _np.inf
                    cleaned = where_device(mask, repl, a)
                    if cleaned is not NotImplemented:
                        return (cleaned.argmax()
                                if func is _np.nanargmax
                                else cleaned.argmin())
        return func(args[0].materialize())
    _nan_kinds = {
        _np.nansum: "sum", _np.nanmean: "mean", _np.nanmax: "max",
        _np.nanmin: "min", _np.nanstd: "std", _np.nanvar: "var",
    }
    if func in _nan_kinds and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"ddof"}:
        kind = _nan_kinds[func]
        ddof = kwargs.get("ddof", 0) if kind in ("std", "var") else 0
        if not kwargs or kind in ("std", "var"):
            r = nan_reduce_device(args[0], kind, ddof)
            if r is not None:
                return r
        return func(args[0].materialize(), **kwargs)
    if func in (_np.cov, _np.corrcoef) and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        r = (cov_device if func is _np.cov else corrcoef_device)(args[0])
        if r is not None:
            return r
        return func(args[0].materialize())
    if func is _np.histogram and 1 <= len(args) <= 2 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"bins", "range"}:
        a = args[0]
        bins = args[1] if len(args) == 2 else kwargs.get("bins", 10)
        rng = kwargs.get("range")
        code = _dtype_code(a.dtype)
        if (
            code is not None
            and a.size >= 1
            and isinstance(bins, (int, _np.integer))
            and 1 <= int(bins) <= 4096
        ):
            bins = int(bins)
            if rng is not None:
                lo, hi = float(rng[0]), float(rng[1])
            else:
                lo, hi = float(a.min()), float(a.max())
            if _np.isfinite(lo) and _np.isfinite(hi) and hi >= lo:
                if hi == lo:  # numpy expands degenerate ranges
                    lo, hi = lo - 0.5, hi + 0.5
                raw = backend().histogram(
                    a._dev_handle(), code, a.size, lo, hi, bins, 1
                )
                counts = _np.frombuffer(raw, dtype=_np.uint64)
                hist = counts[:bins].astype(_np.int64)
                edges = _np.linspace(lo, hi, bins + 1)
                return hist, edges
        host = a.materialize()
        return _np.histogram(host, *args[1:], **kwargs)
    if func in (_np.round, _np.around) and len(args) >= 1 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"decimals"}:
        dec = args[1] if len(args) > 1 else kwargs.get("decimals", 0)
        return args[0].round(dec)
    return _AF_PASS

def _af_structure(func, args, kwargs):
    if func in (_np.concatenate, _np.vstack, _np.hstack, _np.stack) \
            and len(args) == 1 and isinstance(args[0], (list, tuple)):
        axis = kwargs.get("axis", 0)
        parts = args[0]
        routable = (
            set(kwargs) <= {"axis"}
            and axis in (0, None)
            and len(parts) >= 1
            and all(isinstance(p, DeviceArray) for p in parts)
            and len({p.dtype for p in parts}) == 1
            and _dtype_code(parts[0].dtype) is not None
            and all(p._host is None for p in parts)
        )
        if routable:
            shapes = [p.shape for p in parts]
            nd = len(shapes[0])
            if func is _np.concatenate and axis is None:
                out_shape = (sum(p.size for p in parts),)
            elif func in (_np.concatenate, _np.vstack):
                base = shapes[0][1:] if nd > 0 else ()
                if func is _np.vstack and nd == 1:
                    # vstack of 1-D rows -> (k, n)
                    if len({sh for sh in shapes}) == 1:
                        out_shape = (len(parts), shapes[0][0])
                    else:
                        out_shape = None
                elif all(sh[1:] == base for sh in shapes):
                    out_shape = (sum(sh[0] for sh in shapes),) + base
                else:
                    out_shape = None
            elif func is _np.hstack and nd == 1:
                out_shape = (sum(sh[0] for sh in shapes),)
            elif func is _np.stack and len(set(shapes)) == 1:
                out_shape = (len(parts),) + shapes[0]
            else:
                out_shape = None
            if out_shape is not None:
                esz = parts[0].dtype.itemsize
                total = 1
                for d in out_shape:
                    total *= d
                hd = backend().alloc(total * esz)
                out = DeviceArray(hd, out_shape, parts[0].dtype)
                off = 0
                for part in parts:
                    backend().copy_d2d(
                        hd, off, part._dev_handle(), 0,
                        part.size * esz,
                    )
                    off += part.size * esz
                return out
        host = [
            p.materialize() if isinstance(p, DeviceArray) else p
            for p in parts
        ]
        return func(host, **kwargs)
    if func in (_np.reshape, _np.ravel) and len(args) >= 1 and isinstance(
        args[0], DeviceArray
    ) and not kwargs:
        if func is _np.ravel and len(args) == 1:
            return args[0].ravel()
        if func is _np.reshape and len(args) == 2:
            return args[0].reshape(args[1])
    if func is _np.transpose and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ) and set(kwargs) <= {"axes"}:
        return args[0].transpose(*(
            (kwargs["axes"],) if kwargs.get("axes") is not None else ()
        ))
    if func is _np.linalg.norm and len(args) == 1 and isinstance(
        args[0], DeviceArray
    ):
        a = args[0]
        ord_ = kwargs.get("ord")
        axis = kwargs.get("axis")
        code = _dtype_code(a.dtype)
        extra = set(kwargs) - {"ord", "axis"}
        if code is not None and not extra:
            if axis is None and (
                ord_ is None
                or (ord_ == 2 and a.ndim == 1)
                or (ord_ == "fro" and a.ndim == 2)
            ):
                # flat 2-norm / Frobenius: fused square+sum kernel
                return a.dtype.type(float(a.square_sum()) ** 0.5)
            if (
                isinstance(axis, int)
                and ord_ in (None, 2)
                and a.ndim == 2
            ):
                sq = a._unary("square")
                if sq is not NotImplemented:
                    ssum = sq.sum(axis=axis)
                    if isinstance(ssum, DeviceArray):
                        r = ssum._unary("sqrt")
                        if r is not NotImplemented:
                            return r
        host = a.materialize()
        return _np.linalg.norm(host, **kwargs)
    _reductions = {
        _np.max: "max", _np.amax: "max",
        _np.min: "min", _np.amin: "min",
        _np.std: "std", _np.var: "var",
    }
    meth = _reductions.get(func)
    if meth and len(args) == 1 and isinstance(args[0], DeviceArray):
        allowed = {"axis", "ddof", "keepdims"}
        if set(kwargs) <= allowed:
            if kwargs.get("axis") is None and not kwargs.get("keepdims"):
                call_kwargs = {}
                if meth in ("std", "var") and "ddof" in kwargs:
                    call_kwargs["ddof"] = kwargs["ddof"]
                return getattr(args[0], meth)(**call_kwargs)
            if meth in ("max", "min", "std", "var") and kwargs.get(
                "axis"
            ) is not None:
                return getattr(args[0], meth)(**kwargs)
    return _AF_PASS
