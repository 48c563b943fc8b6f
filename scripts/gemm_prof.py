"""GEMM-only loop for PMC profiling (rocprofv3 --pmc ... -- python this)."""

import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "code_interpreter_amd" / "ops"))

import _hipops  # noqa: E402

_hipops.init(0)

which = sys.argv[1] if len(sys.argv) > 1 else "bf16"
size = int(sys.argv[2]) if len(sys.argv) > 2 else 4096

if which == "bf16":
    a = np.random.uniform(-1, 1, (size, size)).astype(np.float32).view(np.uint32)
    a = (a >> 16).astype(np.uint16)
    b = a.copy()
    dt = 2
else:
    dtype = np.float32 if which == "f32" else np.float64
    a = np.random.uniform(-1, 1, (size, size)).astype(dtype)
    b = np.random.uniform(-1, 1, (size, size)).astype(dtype)
    dt = 0 if which == "f32" else 1

ha, hb = _hipops.upload(a), _hipops.upload(b)
import time

for _ in range(3):
    _hipops.free(_hipops.gemm(ha, hb, size, size, size, dt))
_hipops.synchronize()
t0 = time.perf_counter()
iters = 10
for _ in range(iters):
    _hipops.free(_hipops.gemm(ha, hb, size, size, size, dt))
_hipops.synchronize()
dt_s = (time.perf_counter() - t0) / iters
print(f"{which} {size}^3: {dt_s*1e3:.2f} ms  {2*size**3/dt_s/1e12:.1f} TF/s")
