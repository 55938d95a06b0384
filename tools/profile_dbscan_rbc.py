"""rocprofv3 target: ONLY the rbc (ball-cover pruned) DBSCAN fit+transform
on the RESULTS.md 20-blob 1M x 64 shape, so the kernel stats attribute the
sweep time to the pruned path alone."""

from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from spark_rapids_ml_amd import DBSCAN
from spark_rapids_ml_amd.data import DataFrame


def main():
    rng = np.random.default_rng(0)
    n, d = 1_000_000, 64
    C20 = rng.normal(scale=10.0, size=(20, d)).astype(np.float32)
    X = (C20[rng.integers(0, 20, n)]
         + 0.3 * rng.normal(size=(n, d)).astype(np.float32))
    df = DataFrame.from_numpy(X.astype(np.float32))
    torch.cuda.synchronize()
    st = time.perf_counter()
    out = DBSCAN(eps=4.0, min_samples=5, algorithm="rbc").fit(df).transform(df)
    torch.cuda.synchronize()
    lab = np.asarray(out["prediction"])
    print(f"rbc 20blobs: {time.perf_counter() - st:.3f}s "
          f"clusters={int(lab.max()) + 1}", flush=True)


if __name__ == "__main__":
    main()
