"""Full-fit wall-clock sweep at the reference's workload shapes (refreshes
RESULTS.md). One JSON line per row. All timings include ingest + init."""

from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def t0():
    torch.cuda.synchronize()
    return time.perf_counter()


def done(name, start, **kw):
    torch.cuda.synchronize()
    print(json.dumps({"workload": name, "t_s": round(time.perf_counter() - start, 3), **kw}),
          flush=True)


def main():
    from spark_rapids_ml_amd import (
        DBSCAN, KMeans, LinearRegression, LogisticRegression, PCA,
        RandomForestClassifier, RandomForestRegressor, UMAP,
        ApproximateNearestNeighbors, NearestNeighbors,
    )
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    n, d = 1_000_000, 3000
    X = rng.normal(size=(n, d)).astype(np.float32)
    w = rng.normal(size=d).astype(np.float32)
    s = X @ w
    y_cls = (s > 0).astype(np.float64)
    y_reg = (s + 0.1 * rng.normal(size=n)).astype(np.float64)
    df = DataFrame.from_numpy(X)
    df_c = DataFrame.from_numpy(X, y_cls)
    df_r = DataFrame.from_numpy(X, y_reg)

    st = t0(); KMeans(k=1000, maxIter=30, initMode="random", seed=5).fit(df)
    done("kmeans_k1000_iter30", st, ref_2xA10G_s=82)
    st = t0(); PCA(k=3).fit(df); done("pca_k3", st, ref_2xA10G_s=37)
    st = t0(); LinearRegression().fit(df_r); done("linreg_ols", st, ref_2xA10G_s=41)
    st = t0(); LinearRegression(regParam=1e-5).fit(df_r)
    done("ridge", st, ref_2xA10G_s=32)
    st = t0(); LinearRegression(regParam=1e-5, elasticNetParam=0.5, maxIter=10).fit(df_r)
    done("elasticnet_10it", st, ref_2xA10G_s=79)
    st = t0(); LogisticRegression(regParam=1e-5).fit(df_c)
    done("logreg", st, ref_2xA10G_s=69)
    st = t0(); RandomForestClassifier(numTrees=50, maxDepth=13, maxBins=128, seed=1).fit(df_c)
    done("rfc_50t_d13", st, ref_2xA10G_s=59)
    st = t0(); m = RandomForestRegressor(numTrees=30, maxDepth=6, maxBins=128, seed=1).fit(df_r)
    done("rfr_30t_d6", st, ref_2xA10G_s=52)
    # forest transform (fil kernel path) over the full 1M
    st = t0(); m.transform(df_r); done("rfr_transform_1m", st)
    del df_c, df_r, y_cls, y_reg, s, X
    torch.cuda.empty_cache()

    # kNN / ANN / UMAP / DBSCAN shapes from BASELINE.json + RESULTS.md
    Xi = rng.normal(size=(1_000_000, 768)).astype(np.float32)
    Q = rng.normal(size=(10_000, 768)).astype(np.float32)
    nn = NearestNeighbors(k=64).fit(DataFrame.from_numpy(Xi))
    st = t0(); nn.kneighbors(DataFrame.from_numpy(Q)); done("knn_1m_768_k64_10kq", st)
    del Xi
    torch.cuda.empty_cache()

    X10 = rng.normal(size=(10_000_000, 768)).astype(np.float32)
    nn = NearestNeighbors(k=64).fit(DataFrame.from_numpy(X10))
    st = t0(); nn.kneighbors(DataFrame.from_numpy(Q)); done("knn_10m_768_k64_10kq", st, r01_s=5.62)
    del X10, nn
    torch.cuda.empty_cache()

    Xa = rng.normal(size=(1_000_000, 256)).astype(np.float32)
    dfa = DataFrame.from_numpy(Xa)
    st = t0()
    ann = ApproximateNearestNeighbors(k=64, algorithm="ivfflat").fit(dfa)
    ann.kneighbors(DataFrame.from_numpy(Xa[:10_000]))
    done("ivfflat_build_search_1m_256", st, r01_s=1.01)

    st = t0(); UMAP(n_epochs=200).fit(dfa); done("umap_1m_256_200ep", st, r01_s=11.8)
    del Xa, dfa
    torch.cuda.empty_cache()

    Xd = rng.normal(size=(1_000_000, 64)).astype(np.float32)
    dfd = DataFrame.from_numpy(Xd)
    st = t0(); DBSCAN(eps=1.0, min_samples=5).fit(dfd).transform(dfd)
    done("dbscan_1m_64_allnoise", st, r01_s=4.94)
    # clustered data: label propagation actually runs
    C20 = rng.normal(scale=10.0, size=(20, 64)).astype(np.float32)
    Xc = (C20[rng.integers(0, 20, 1_000_000)]
          + 0.3 * rng.normal(size=(1_000_000, 64)).astype(np.float32))
    dfc2 = DataFrame.from_numpy(Xc.astype(np.float32))
    st = t0()
    out = DBSCAN(eps=4.0, min_samples=5).fit(dfc2).transform(dfc2)
    done("dbscan_1m_64_20blobs", st,
         n_clusters=int(np.asarray(out["prediction"]).max()) + 1)
    st = t0()
    out = DBSCAN(eps=4.0, min_samples=5, algorithm="rbc").fit(dfc2).transform(dfc2)
    done("dbscan_1m_64_20blobs_rbc", st,
         n_clusters=int(np.asarray(out["prediction"]).max()) + 1)

    # sparse logreg 50M x 2048 @ 1% (BASELINE.json config)
    del Xd
    torch.cuda.empty_cache()
    from benchmark.gen_data import gen_sparse_classification_fast

    Xs, ys = gen_sparse_classification_fast(50_000_000, 2048, nnz_per_row=20, seed=0)
    dfs = DataFrame({"features": Xs, "label": ys})
    st = t0(); LogisticRegression(regParam=1e-5, maxIter=30).fit(dfs)
    done("sparse_logreg_50m_2048_1g_nnz", st, r01_s=11.7)


if __name__ == "__main__":
    main()
