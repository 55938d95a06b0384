"""Short RFR fit for rocprof attribution (10 trees, headline shape)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from spark_rapids_ml_amd import RandomForestRegressor
from spark_rapids_ml_amd.data import DataFrame

rng = np.random.default_rng(0)
n, d = 1_000_000, 3000
X = rng.normal(size=(n, d)).astype(np.float32)
y = (X @ rng.normal(size=d).astype(np.float32)).astype(np.float64)
df = DataFrame.from_numpy(X, y)
t0 = time.perf_counter()
RandomForestRegressor(numTrees=10, maxDepth=6, maxBins=128, seed=1).fit(df)
import torch; torch.cuda.synchronize()
print(f"rfr 10-tree fit: {time.perf_counter()-t0:.2f}s", flush=True)
