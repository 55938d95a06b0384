"""ANN recall tests (pattern: reference
tests/test_approximate_nearest_neighbors.py)."""

import numpy as np
import pytest
from sklearn.neighbors import NearestNeighbors as SkNN

from spark_rapids_ml_amd import ApproximateNearestNeighbors
from spark_rapids_ml_amd.data import DataFrame


def _data(n=2000, d=16, seed=0):
    rng = np.random.default_rng(seed)
    return rng.normal(size=(n, d)).astype(np.float32)


def _recall(idx, sk_idx):
    hits = 0
    for a, b in zip(idx, sk_idx):
        hits += len(set(a.tolist()) & set(b.tolist()))
    return hits / sk_idx.size


@pytest.mark.parametrize("algo,params", [
    ("ivfflat", {"nlist": 32, "nprobe": 8}),
    ("ivfpq", {"nlist": 32, "nprobe": 8, "refine_ratio": 4.0}),
])
def test_ann_recall(algo, params):
    X = _data()
    Q = X[:100]
    model = ApproximateNearestNeighbors(k=10, algorithm=algo, algoParams=params).fit(
        DataFrame.from_numpy(X)
    )
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(Q))
    idx = np.asarray(knn_df["indices"])
    sk = SkNN(n_neighbors=10).fit(X)
    _, sk_idx = sk.kneighbors(Q)
    rec = _recall(idx, sk_idx)
    assert rec > 0.7, f"{algo} recall {rec}"


def test_ann_full_probe_is_exact():
    X = _data(n=500)
    Q = X[:50]
    model = ApproximateNearestNeighbors(
        k=5, algorithm="ivfflat", algoParams={"nlist": 8, "nprobe": 8}
    ).fit(DataFrame.from_numpy(X))
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(Q))
    idx = np.asarray(knn_df["indices"])
    sk = SkNN(n_neighbors=5).fit(X)
    _, sk_idx = sk.kneighbors(Q)
    assert _recall(idx, sk_idx) == 1.0


def test_ann_unsupported_algo_raises():
    X = _data(n=100)
    model = ApproximateNearestNeighbors(k=3, algorithm="bogus").fit(
        DataFrame.from_numpy(X)
    )
    with pytest.raises(ValueError):
        model.kneighbors(DataFrame.from_numpy(X[:5]))


def test_cagra_recall():
    X = _data(n=3000, d=16)
    Q = X[:100]
    model = ApproximateNearestNeighbors(
        k=10,
        algorithm="cagra",
        algoParams={"graph_degree": 32, "itopk_size": 128, "max_iterations": 10},
    ).fit(DataFrame.from_numpy(X))
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(Q))
    idx = np.asarray(knn_df["indices"])
    from sklearn.neighbors import NearestNeighbors as SkNN

    sk = SkNN(n_neighbors=10).fit(X)
    _, sk_idx = sk.kneighbors(Q)
    assert _recall(idx, sk_idx) > 0.6


def _dist_ann(algo):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(600, 16)).astype(np.float32)
    Q = rng.normal(size=(30, 16)).astype(np.float32)
    ids = np.arange(600, dtype=np.int64)
    df = DataFrame({
        "features": X[comm.rank :: comm.world_size],
        "id": ids[comm.rank :: comm.world_size],
    })
    qdf = DataFrame({"features": Q[comm.rank :: comm.world_size]})
    model = ApproximateNearestNeighbors(
        k=8, algorithm=algo, algoParams={"nlist": 8, "nprobe": 8}, idCol="id"
    ).fit(df)
    _, _, knn_df = model.kneighbors(qdf)
    return np.asarray(knn_df["indices"]), np.asarray(knn_df["distances"])


def test_ann_distributed_recall():
    """Each rank searches its local index over the replicated queries and the
    partial top-k merge assembles global neighbors (reference ANN is
    comms-free per partition + Spark groupBy merge, knn.py:1282-1322)."""
    from sklearn.neighbors import NearestNeighbors as SkNN

    from tests.dist_utils import run_distributed

    results = run_distributed(_dist_ann, world_size=2, args=("ivfflat",))
    rng = np.random.default_rng(0)
    X = rng.normal(size=(600, 16)).astype(np.float32)
    Q = rng.normal(size=(30, 16)).astype(np.float32)
    _, sk_idx = SkNN(n_neighbors=8).fit(X).kneighbors(Q)
    hits = total = 0
    for r, (idx, _d) in enumerate(results):
        truth = sk_idx[r::2]
        for row_pred, row_true in zip(idx, truth):
            hits += len(set(row_pred) & set(row_true))
            total += len(row_true)
    assert hits / total > 0.85, f"recall {hits/total}"


def test_approx_similarity_join():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(2000, 32)).astype(np.float32)
    Q = rng.normal(size=(50, 32)).astype(np.float32)
    model = ApproximateNearestNeighbors(k=4).fit(DataFrame.from_numpy(X))
    joined = model.approxSimilarityJoin(DataFrame.from_numpy(Q), distCol="d")
    assert joined.num_rows == 50 * 4
    assert "d" in joined.columns
    assert float(np.asarray(joined["d"]).min()) >= 0.0
