# Preinstalled science stack: numpy + pandas + scipy in one payload.
# Any import that is NOT preinstalled triggers the executor's AST scan
# and wheelhouse pip install before the script runs.
import numpy as np
import pandas as pd
from scipy import stats

rng = np.random.default_rng(7)
a = rng.normal(5.0, 1.0, 200)
b = rng.normal(5.4, 1.0, 200)
frame = pd.DataFrame({"a": a, "b": b})
print(frame.describe().loc[["mean", "std"]].round(3))
t, p = stats.ttest_ind(a, b)
print("t =", round(float(t), 3), " p =", round(float(p), 6))
