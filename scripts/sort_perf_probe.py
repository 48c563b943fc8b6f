"""Same-box timing of the device radix sort vs np.sort on host, 1e7/1e8."""
import sys, time
from pathlib import Path
import numpy as np
OPS = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"
sys.path.insert(0, str(OPS))
import _hipops as hip
hip.init(0)

for dtype, code, n in [(np.float32, 0, 10_000_000), (np.float64, 1, 10_000_000),
                       (np.float32, 0, 100_000_000)]:
    a = np.random.default_rng(0).standard_normal(n).astype(dtype)
    h = hip.upload(a)
    # warm
    hs = hip.sort(h, code, n, 0); hip.free(hs); hip.synchronize()
    ts = []
    for _ in range(5):
        t0 = time.perf_counter()
        hs = hip.sort(h, code, n, 0)
        hip.synchronize()
        ts.append(time.perf_counter() - t0)
        hip.free(hs)
    dev = min(ts)
    t0 = time.perf_counter(); np.sort(a); host = time.perf_counter() - t0
    print(f"{np.dtype(dtype).name} n={n:>11,}: device {dev*1e3:8.2f} ms "
          f"({n/dev/1e9:6.2f} Gelem/s)  np.sort {host*1e3:8.2f} ms  "
          f"speedup {host/dev:6.1f}x")
    hip.free(h)
