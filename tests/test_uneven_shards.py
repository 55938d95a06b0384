"""Distributed correctness with uneven/empty rank shards (an 8-GPU run on a
small dataset leaves ranks empty; every collective path must stay aligned)."""

import numpy as np
import pytest

from spark_rapids_ml_amd import KMeans, LinearRegression, LogisticRegression, PCA
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _dist_empty_shard(algo: str):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 8))
    w = rng.normal(size=8)
    y = X @ w + 0.01 * rng.normal(size=200)
    # rank 1 gets an EMPTY shard
    if comm.rank == 0:
        Xl, yl = X, y
    else:
        Xl, yl = X[:0], y[:0]
    if algo == "kmeans":
        m = KMeans(k=3, maxIter=10, seed=1).fit(DataFrame.from_numpy(Xl.astype(np.float32)))
        return m.cluster_centers_
    if algo == "linreg":
        m = LinearRegression().fit(DataFrame.from_numpy(Xl, yl))
        return np.asarray(m.coefficients)
    if algo == "logreg":
        m = LogisticRegression(maxIter=50).fit(
            DataFrame.from_numpy(Xl, (yl > 0).astype(np.float64))
        )
        return np.asarray(m.coefficients)
    if algo == "pca":
        m = PCA(k=2).fit(DataFrame.from_numpy(Xl))
        return np.asarray(m.components_)
    raise ValueError(algo)


@pytest.mark.parametrize("algo", ["kmeans", "linreg", "logreg", "pca"])
def test_empty_shard_rank(algo):
    results = run_distributed(_dist_empty_shard, world_size=2, args=(algo,))
    assert np.allclose(results[0], results[1], atol=1e-5)
