"""Raw kernel performance probe (run on the GPU box):
bandwidth + TFLOP/s for each hot kernel, printed one line per kernel.
"""

import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "code_interpreter_amd" / "ops"))

import _hipops  # noqa: E402

_hipops.init(0)


def timed(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    _hipops.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    _hipops.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    n = 10**8
    # rand f64: writes 8n bytes
    h = _hipops.rand(n, 1, 1)
    t = timed(lambda: _hipops.free(_hipops.rand(n, 1, 2)))
    print(f"rand_f64      n=1e8: {t*1e3:7.2f} ms  {8*n/t/1e12:6.2f} TB/s (write)")

    # square f64: 16n bytes traffic
    t = timed(lambda: _hipops.free(_hipops.unary(h, 0, 1, n)))
    print(f"square_f64    n=1e8: {t*1e3:7.2f} ms  {16*n/t/1e12:6.2f} TB/s (r+w)")

    # sum f64 (8n read)
    t = timed(lambda: _hipops.sum(h, 1, n, 0))
    print(f"sum_f64       n=1e8: {t*1e3:7.2f} ms  {8*n/t/1e12:6.2f} TB/s (read)")

    # fused square+sum
    t = timed(lambda: _hipops.sum(h, 1, n, 1))
    print(f"sqsum_f64     n=1e8: {t*1e3:7.2f} ms  {8*n/t/1e12:6.2f} TB/s (read)")

    # max reduction (8n read, NaN-propagating)
    t = timed(lambda: _hipops.sum(h, 1, n, 2))
    print(f"max_f64       n=1e8: {t*1e3:7.2f} ms  {8*n/t/1e12:6.2f} TB/s (read)")

    # transcendental unary (log): 16n bytes traffic
    t = timed(lambda: _hipops.free(_hipops.unary(h, 5, 1, n)))
    print(f"log_f64       n=1e8: {t*1e3:7.2f} ms  {16*n/t/1e12:6.2f} TB/s (r+w)")
    _hipops.free(h)

    # normal RNG (Box-Muller, fused affine): writes 8n bytes
    t = timed(lambda: _hipops.free(_hipops.randn(n, 3, 0.0, 1.0)))
    print(f"randn_f64     n=1e8: {t*1e3:7.2f} ms  {8*n/t/1e12:6.2f} TB/s (write)")

    # upload 800 MB (pinned staged)
    host = np.random.rand(n)
    t = timed(lambda: _hipops.free(_hipops.upload(host)), iters=3, warmup=1)
    print(f"upload        800MB: {t*1e3:7.2f} ms  {8*n/t/1e9:6.1f} GB/s (PCIe)")

    # GEMMs
    for name, dt, size, esz in (
        ("gemm_f32", 0, 4096, 4),
        ("gemm_f64", 1, 2048, 8),
        ("gemm_bf16", 2, 4096, 2),
    ):
        if dt == 2:
            a = (np.random.uniform(-1, 1, (size, size)).astype(np.float32)).view(
                np.uint32
            )
            a = ((a >> 16).astype(np.uint16))
            b = a.copy()
        else:
            dtype = np.float32 if dt == 0 else np.float64
            a = np.random.uniform(-1, 1, (size, size)).astype(dtype)
            b = np.random.uniform(-1, 1, (size, size)).astype(dtype)
        ha, hb = _hipops.upload(a), _hipops.upload(b)
        t = timed(lambda: _hipops.free(_hipops.gemm(ha, hb, size, size, size, dt)),
                  iters=5, warmup=2)
        tf = 2 * size**3 / t / 1e12
        print(f"{name:13s} {size}^3: {t*1e3:7.2f} ms  {tf:7.1f} TF/s")
        _hipops.free(ha)
        _hipops.free(hb)


if __name__ == "__main__":
    main()
