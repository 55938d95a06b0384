"""Exact kNN (pattern: reference tests/test_nearest_neighbors.py)."""

import numpy as np
import pytest
from sklearn.neighbors import NearestNeighbors as SkNN

from spark_rapids_ml_amd import NearestNeighbors
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _data(n=300, d=8, seed=0):
    rng = np.random.default_rng(seed)
    return rng.normal(size=(n, d)).astype(np.float32)


def test_exact_knn_matches_sklearn():
    X = _data()
    Q = _data(n=50, seed=1)
    model = NearestNeighbors(k=5).fit(DataFrame.from_numpy(X))
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(Q))
    idx = np.asarray(knn_df["indices"])
    dist = np.asarray(knn_df["distances"])
    sk = SkNN(n_neighbors=5).fit(X)
    sk_dist, sk_idx = sk.kneighbors(Q)
    assert np.array_equal(idx, sk_idx)
    assert np.allclose(dist, sk_dist, atol=1e-4)


def test_knn_join_explodes():
    X = _data(n=100)
    Q = _data(n=10, seed=1)
    model = NearestNeighbors(k=3).fit(DataFrame.from_numpy(X))
    joined = model.exactNearestNeighborsJoin(DataFrame.from_numpy(Q), distCol="dist")
    assert joined.num_rows == 30
    assert "dist" in joined.columns


def test_knn_custom_id_col():
    X = _data(n=50)
    ids = np.arange(1000, 1050, dtype=np.int64)
    df = DataFrame({"features": X, "row_id": ids})
    model = NearestNeighbors(k=3, idCol="row_id").fit(df)
    _, _, knn_df = model.kneighbors(df)
    idx = np.asarray(knn_df["indices"])
    assert idx.min() >= 1000
    # nearest neighbor of a point is itself
    assert np.array_equal(idx[:, 0], ids)


def _dist_knn(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X = _data(n=300, seed=seed)
    Q = _data(n=40, seed=seed + 1)
    ids = np.arange(len(X), dtype=np.int64)
    df = DataFrame({"features": X[comm.rank :: comm.world_size], "id": ids[comm.rank :: comm.world_size]})
    qdf = DataFrame({"features": Q[comm.rank :: comm.world_size]})
    model = NearestNeighbors(k=4, idCol="id").fit(df)
    _, _, knn_df = model.kneighbors(qdf)
    return np.asarray(knn_df["indices"]), np.asarray(knn_df["distances"])


def test_knn_distributed_matches_single():
    results = run_distributed(_dist_knn, world_size=2, args=(0,))
    X = _data(n=300, seed=0)
    Q = _data(n=40, seed=1)
    sk = SkNN(n_neighbors=4).fit(X)
    sk_dist, sk_idx = sk.kneighbors(Q)
    # reassemble: rank r got queries r::2
    for r, (idx, dist) in enumerate(results):
        assert np.array_equal(idx, sk_idx[r::2])
        assert np.allclose(dist, sk_dist[r::2], atol=1e-4)


def test_knn_model_persistence_unsupported():
    X = _data(n=50)
    model = NearestNeighbors(k=3).fit(DataFrame.from_numpy(X))
    with pytest.raises(NotImplementedError):
        model.save("/tmp/should_not_exist")


def test_approx_similarity_join_alias():
    X = _data(n=80)
    Q = _data(n=8, seed=1)
    model = NearestNeighbors(k=3).fit(DataFrame.from_numpy(X))
    joined = model.approxSimilarityJoin(DataFrame.from_numpy(Q))
    assert joined.num_rows == 24


def test_knn_k_exceeds_items_pads_with_sentinels():
    X = _data(n=60, d=4)
    _, _, knn = NearestNeighbors(k=100).fit(DataFrame.from_numpy(X)).kneighbors(
        DataFrame.from_numpy(X[:3])
    )
    idx = np.asarray(knn["indices"])
    dist = np.asarray(knn["distances"])
    assert idx.shape == (3, 100)
    assert (idx[:, :60] >= 0).all()
    assert (idx[:, 60:] == -1).all()
    assert np.isinf(dist[:, 60:]).all()
