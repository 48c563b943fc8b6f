"""Data-analytics surface demo: on a GPU engine every step below runs on
the gfx950 device kernels (sort, quantiles, histogram, covariance); on a
CPU-only deployment the same code runs stock numpy - identical values.
"""
import numpy as np

rng_rows, rng_cols = 400, 5000
x = np.random.rand(rng_rows, rng_cols)

row_med = np.median(x, axis=1)
q = np.quantile(x, [0.05, 0.5, 0.95])
hist, edges = np.histogram(x, bins=20)
top = np.sort(x[0])
order = np.argsort(x[0])
corr = np.corrcoef(x[:16])
cum = np.cumsum(x, axis=1)
z = (x - x.mean(axis=1, keepdims=True)) / x.std()

print("row-median mean:", round(float(row_med.mean()), 3))
print("quantiles:", [round(float(v), 3) for v in np.asarray(q)])
print("hist total:", int(hist.sum()), "edges:", round(float(edges[0]), 2),
      round(float(edges[-1]), 2))
print("sorted head ascending:", bool(np.all(np.asarray(top)[:5][:-1]
                                            <= np.asarray(top)[:5][1:])))
print("argsort first is min:", int(order[0]) == int(np.argmin(x[0])))
print("corr diag ~1:", bool(abs(float(np.asarray(corr)[0, 0]) - 1.0) < 1e-9))
print("cumsum last col == row sums:",
      bool(np.allclose(np.asarray(cum)[:, -1], np.asarray(x.sum(axis=1)),
                       rtol=1e-8)))
print("z-score mean ~0:", bool(abs(float(z.mean())) < 1e-6))
