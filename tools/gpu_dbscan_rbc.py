"""Validate + time DBSCAN algorithm="rbc" (ball-cover tile pruning) against
the dense sweep on the RESULTS.md clustered-data shape (1M x 64, 20 blobs)."""

from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from spark_rapids_ml_amd import DBSCAN
from spark_rapids_ml_amd.data import DataFrame
from sklearn.metrics import adjusted_rand_score


def timed(tag, est, df):
    torch.cuda.synchronize()
    st = time.perf_counter()
    out = est.fit(df).transform(df)
    torch.cuda.synchronize()
    lab = np.asarray(out["prediction"])
    print(json.dumps({
        "case": tag, "t_s": round(time.perf_counter() - st, 3),
        "n_clusters": int(lab.max()) + 1, "noise": int((lab == -1).sum()),
    }), flush=True)
    return lab


def main():
    rng = np.random.default_rng(0)
    n, d = 1_000_000, 64
    C20 = rng.normal(scale=10.0, size=(20, d)).astype(np.float32)
    X = (C20[rng.integers(0, 20, n)]
         + 0.3 * rng.normal(size=(n, d)).astype(np.float32))
    df = DataFrame.from_numpy(X.astype(np.float32))

    b = timed("brute_20blobs", DBSCAN(eps=4.0, min_samples=5), df)
    r = timed("rbc_20blobs", DBSCAN(eps=4.0, min_samples=5, algorithm="rbc"), df)
    ars = adjusted_rand_score(b, r)
    same_noise = bool(np.array_equal(b == -1, r == -1))
    print(json.dumps({"ars_brute_vs_rbc": ars, "same_noise": same_noise}), flush=True)
    assert ars == 1.0 and same_noise

    # uniform noise: the ball bound prunes nothing -> should fall back dense
    Xn = rng.normal(size=(300_000, d)).astype(np.float32)
    dfn = DataFrame.from_numpy(Xn)
    bn = timed("brute_noise300k", DBSCAN(eps=1.0, min_samples=5), dfn)
    rn = timed("rbc_noise300k", DBSCAN(eps=1.0, min_samples=5, algorithm="rbc"), dfn)
    assert np.array_equal(bn, rn)
    print("OK", flush=True)


if __name__ == "__main__":
    main()
