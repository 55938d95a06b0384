"""Round-2 kernel A/B microbench: new HIP paths vs round-1 defaults.

Run on a GPU box:  python tools/bench_r02.py knn|rf|umap|all
Writes one JSON line per measurement to stdout.
"""

from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def _sync():
    torch.cuda.synchronize()


def _t(fn, warm=1, rep=3):
    for _ in range(warm):
        fn()
    _sync()
    t0 = time.perf_counter()
    for _ in range(rep):
        fn()
    _sync()
    return (time.perf_counter() - t0) / rep


def bench_knn():
    from spark_rapids_ml_amd.ops import torch_ref
    from spark_rapids_ml_amd.ops.knn import _knn_topk_gemm_select

    for nq, ni, d, k in [(10000, 1_000_000, 768, 64), (10000, 100_000, 768, 64)]:
        g = torch.Generator(device="cuda").manual_seed(0)
        Q = torch.randn(nq, d, generator=g, device="cuda")
        I = torch.randn(ni, d, generator=g, device="cuda")
        t_torch = _t(lambda: torch_ref.knn_topk(Q, I, k), warm=1, rep=2)
        t_new = _t(lambda: _knn_topk_gemm_select(Q, I, k), warm=1, rep=2)
        # correctness spot check
        dd, ii = _knn_topk_gemm_select(Q, I, k)
        rd, ri = torch_ref.knn_topk(Q, I, k)
        ok = bool(torch.allclose(dd, rd, rtol=1e-3, atol=1e-2))
        print(json.dumps({"bench": "knn_topk", "nq": nq, "ni": ni, "d": d, "k": k,
                          "t_torch_s": round(t_torch, 4), "t_new_s": round(t_new, 4),
                          "speedup": round(t_torch / t_new, 2), "match": ok}), flush=True)
        del Q, I
        torch.cuda.empty_cache()


def bench_rf():
    from spark_rapids_ml_amd import RandomForestClassifier, RandomForestRegressor
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    n, d = 1_000_000, 3000
    X = rng.normal(size=(n, d)).astype(np.float32)
    w = rng.normal(size=d).astype(np.float32)
    s = X @ w
    y_cls = (s > 0).astype(np.float64)
    y_reg = (s + 0.1 * rng.normal(size=n)).astype(np.float64)

    df_c = DataFrame.from_numpy(X, y_cls)
    t0 = time.perf_counter()
    RandomForestClassifier(numTrees=50, maxDepth=13, maxBins=128, seed=1).fit(df_c)
    _sync()
    t_rfc = time.perf_counter() - t0
    print(json.dumps({"bench": "rfc_fit", "n": n, "d": d, "trees": 50, "depth": 13,
                      "t_s": round(t_rfc, 2), "r01_was_s": 2.3}), flush=True)

    df_r = DataFrame.from_numpy(X, y_reg)
    t0 = time.perf_counter()
    RandomForestRegressor(numTrees=30, maxDepth=6, maxBins=128, seed=1).fit(df_r)
    _sync()
    t_rfr = time.perf_counter() - t0
    print(json.dumps({"bench": "rfr_fit", "n": n, "d": d, "trees": 30, "depth": 6,
                      "t_s": round(t_rfr, 2), "r01_was_s": 3.8}), flush=True)


def bench_umap():
    from spark_rapids_ml_amd import UMAP
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    X = rng.normal(size=(1_000_000, 256)).astype(np.float32)
    df = DataFrame.from_numpy(X)
    t0 = time.perf_counter()
    UMAP(n_epochs=200).fit(df)
    _sync()
    t = time.perf_counter() - t0
    print(json.dumps({"bench": "umap_fit_1m", "t_s": round(t, 2), "r01_was_s": 11.8}),
          flush=True)


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("knn", "all"):
        bench_knn()
    if which in ("rf", "all"):
        bench_rf()
    if which in ("umap", "all"):
        bench_umap()


if __name__ == "__main__":
    main()
