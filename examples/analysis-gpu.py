# r02 device-surface showcase: a realistic analysis chain where every
# step stays GPU-resident (no 800 MB host round-trips): RNG, axis stats,
# broadcasting, boolean masks, selection, quantiles.
import numpy as np
import time

t0 = time.time()
x = np.random.rand(5000, 4000)             # Philox on device
col_mean = x.mean(axis=0)                  # axis reduction
centered = x - col_mean                    # broadcast subtract
outliers = centered > 0.45                 # device boolean mask
n_out = int(outliers.sum())                # popcount
trimmed = np.where(outliers, 0.0, centered)
spread = trimmed.std(axis=1)               # composed axis var/std
p95 = float(np.percentile(trimmed, 95.0))  # histogram-bisection select
peak_row = int(np.argmax(spread))
total = float(np.sum(np.square(trimmed)))  # fused square+sum
dt = time.time() - t0
print(f"kind: {type(centered).__name__}")
print(f"outliers: {n_out}  p95: {p95:.4f}  peak_row: {peak_row}")
print(f"sum sq: {total:.3f}  elapsed: {dt*1000:.1f} ms")
