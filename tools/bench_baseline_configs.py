"""The remaining BASELINE.json named configs, measured for the record."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import json
import numpy as np, torch

which = sys.argv[1] if len(sys.argv) > 1 else "all"

def sync(): torch.cuda.synchronize()

if which in ("rf500", "all"):
    from spark_rapids_ml_amd import RandomForestClassifier
    from spark_rapids_ml_amd.data import DataFrame
    from benchmark.gen_data import gen_classification
    X, y = gen_classification(20_000_000, 256, n_classes=2, n_informative=64, seed=0)
    df = DataFrame.from_numpy(X, y)
    t0 = time.perf_counter()
    m = RandomForestClassifier(numTrees=500, maxDepth=16, maxBins=128, seed=1).fit(df)
    sync()
    t = time.perf_counter() - t0
    out = m.transform(DataFrame.from_numpy(X[:1_000_000]))
    acc = float((np.asarray(out["prediction"]) == y[:1_000_000]).mean())
    print(json.dumps({"bench": "rfc_500t_d16_20m_256", "t_s": round(t, 2),
                      "acc_train_1m": round(acc, 4)}), flush=True)
    del X, y, df
    torch.cuda.empty_cache()

if which in ("km100m", "all"):
    from spark_rapids_ml_amd.ops import kmeans_assign_reduce
    g = torch.Generator(device="cuda").manual_seed(0)
    n, d, k = 100_000_000, 128, 200
    X = torch.randn(n, d, generator=g, device="cuda")
    C = torch.randn(k, d, generator=g, device="cuda")
    x_sq = (X * X).sum(dim=1)
    kmeans_assign_reduce(X, C, x_sq); sync()
    t0 = time.perf_counter()
    for _ in range(5):
        labels, sums, counts, inertia = kmeans_assign_reduce(X, C, x_sq)
    sync()
    t = (time.perf_counter() - t0) / 5
    print(json.dumps({"bench": "kmeans_100m_128_k200_step", "ms_per_step": round(t*1000, 1),
                      "r01_ms": 261}), flush=True)
