"""Rectangular/odd-shape GEMM numerics across dtypes (tail-guard paths)."""
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"))
import _hipops

_hipops.init(0)

def to_bf16(a):
    u = a.astype(np.float32).view(np.uint32)
    return ((u + 0x7FFF + ((u >> 16) & 1)) >> 16).astype(np.uint16)

def from_bf16(b):
    return (b.astype(np.uint32) << 16).view(np.float32)

SHAPES = [(100, 300, 200), (257, 129, 65), (512, 128, 384), (1, 1, 1),
          (64, 1024, 32), (333, 77, 513), (256, 512, 128), (768, 256, 640)]
bad = 0
for rep in range(2):
    for (m, n, k) in SHAPES:
        for dt, tol in ((0, 2e-2), (1, 1e-9), (2, 2e-2)):
            rng = np.random.default_rng(rep * 31 + m + n + k + dt)
            if dt == 2:
                a = to_bf16(rng.uniform(-1, 1, (m, k)))
                b = to_bf16(rng.uniform(-1, 1, (k, n)))
            else:
                dtype = np.float32 if dt == 0 else np.float64
                a = rng.uniform(-1, 1, (m, k)).astype(dtype)
                b = rng.uniform(-1, 1, (k, n)).astype(dtype)
            ha, hb = _hipops.upload(a), _hipops.upload(b)
            hc = _hipops.gemm(ha, hb, m, n, k, dt)
            out = np.empty((m, n), dtype=a.dtype)
            _hipops.download(hc, out)
            for h in (ha, hb, hc):
                _hipops.free(h)
            if dt == 2:
                got = from_bf16(out).astype(np.float64)
                ref = from_bf16(a).astype(np.float64) @ from_bf16(b).astype(np.float64)
            else:
                got = out.astype(np.float64)
                ref = a.astype(np.float64) @ b.astype(np.float64)
            err = float(np.max(np.abs(got - ref) / (np.abs(ref) + 1.0)))
            ok = err < tol
            bad += not ok
            if not ok:
                print(f"FAIL rep{rep} dt{dt} {m}x{n}x{k}: {err}")
print("shape screen:", "ALL OK" if bad == 0 else f"{bad} FAILURES")
sys.exit(1 if bad else 0)
