"""Minimal RFR-histogram-only workload for PMC collection."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from spark_rapids_ml_amd import RandomForestRegressor
from spark_rapids_ml_amd.data import DataFrame

rng = np.random.default_rng(0)
n, d = 1_000_000, 3000
X = rng.normal(size=(n, d)).astype(np.float32)
y = (X @ rng.normal(size=d).astype(np.float32)).astype(np.float64)
RandomForestRegressor(numTrees=3, maxDepth=3, maxBins=128, seed=1).fit(DataFrame.from_numpy(X, y))
torch.cuda.synchronize()
print("done")
