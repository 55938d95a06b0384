import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from benchmark.gen_data import gen_sparse_classification_fast
from spark_rapids_ml_amd import LogisticRegression
from spark_rapids_ml_amd.data import DataFrame

Xs, ys = gen_sparse_classification_fast(20_000_000, 2048, nnz_per_row=20, seed=0)
df = DataFrame({"features": Xs, "label": ys})
t0 = time.perf_counter()
m = LogisticRegression(regParam=1e-5, maxIter=30).fit(df)
torch.cuda.synchronize()
print(f"sparse logreg 20M fit: {time.perf_counter()-t0:.2f}s iters={m.numIters if hasattr(m,'numIters') else '?'}")
