import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from spark_rapids_ml_amd import RandomForestClassifier
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.ops.dispatch import has_hip_ops, hip_ops

print("has_hip_ops:", has_hip_ops())
ext = hip_ops()
print("rf_partition present:", hasattr(ext, "rf_partition"))

rng = np.random.default_rng(0)
n, d = 200_000, 100
X = rng.normal(size=(n, d)).astype(np.float32)
y = (X[:, 0] + X[:, 1] > 0).astype(np.float64)
df = DataFrame.from_numpy(X, y)
t0 = time.perf_counter()
m = RandomForestClassifier(numTrees=4, maxDepth=8, maxBins=64, seed=1).fit(df)
torch.cuda.synchronize()
print(f"fit: {time.perf_counter()-t0:.2f}s")
nodes = [t["feature"].shape[0] for t in m.trees]
print("tree node counts:", nodes)
pred = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
print("train acc:", float((pred == y).mean()))
