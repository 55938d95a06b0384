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


def _dist_knn_empty_items(_):
    from spark_rapids_ml_amd import NearestNeighbors
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(100, 8)).astype(np.float32)
    Q = rng.normal(size=(20, 8)).astype(np.float32)
    ids = np.arange(100, dtype=np.int64)
    # rank 1 holds NO items but does hold queries
    if comm.rank == 0:
        df = DataFrame({"features": X, "id": ids})
    else:
        df = DataFrame({"features": X[:0], "id": ids[:0]})
    qdf = DataFrame({"features": Q[comm.rank :: comm.world_size]})
    model = NearestNeighbors(k=4, idCol="id").fit(df)
    _, _, knn_df = model.kneighbors(qdf)
    return np.asarray(knn_df["indices"]), np.asarray(knn_df["distances"])


def test_knn_empty_item_shard():
    from sklearn.neighbors import NearestNeighbors as SkNN

    results = run_distributed(_dist_knn_empty_items, world_size=2, args=(None,))
    rng = np.random.default_rng(0)
    X = rng.normal(size=(100, 8)).astype(np.float32)
    Q = rng.normal(size=(20, 8)).astype(np.float32)
    sk_dist, sk_idx = SkNN(n_neighbors=4).fit(X).kneighbors(Q)
    for r, (idx, dist) in enumerate(results):
        assert np.array_equal(idx, sk_idx[r::2])
        assert np.allclose(dist, sk_dist[r::2], atol=1e-4)


def _dist_rf_empty(_):
    from spark_rapids_ml_amd import RandomForestClassifier

    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    if comm.rank == 0:
        Xl, yl = X, y
    else:
        Xl, yl = X[:0], y[:0]
    m = RandomForestClassifier(numTrees=4, maxDepth=3, seed=1).fit(DataFrame.from_numpy(Xl, yl))
    out = m.transform(DataFrame.from_numpy(X))
    acc = (np.asarray(out["prediction"]) == y).mean()
    return m.numTrees, acc


def test_rf_empty_shard_rank():
    """Trees assigned to a rank with zero local rows must not break the
    ensemble merge (an 8-GPU fit on a small dataset hits this)."""
    results = run_distributed(_dist_rf_empty, world_size=2, args=(None,))
    for n_trees, acc in results:
        assert n_trees == 4
        assert acc > 0.9


def test_three_rank_world():
    """world_size=3 catches modular-arithmetic assumptions a 2-rank world
    hides (odd rank counts happen on partially-allocated nodes)."""
    results = run_distributed(_dist_empty_shard, world_size=3, args=("kmeans",))
    assert np.allclose(results[0], results[1])
    assert np.allclose(results[1], results[2])


def test_empty_transform_all_models():
    """transform of a 0-row shard works for every model family (an 8-GPU
    transform on a small dataset leaves ranks empty)."""
    from sklearn.datasets import make_blobs, make_classification

    from spark_rapids_ml_amd import (
        KMeans,
        LinearRegression,
        LogisticRegression,
        PCA,
        RandomForestClassifier,
        UMAP,
    )

    Xb, _ = make_blobs(n_samples=200, n_features=6, centers=3, random_state=0)
    Xb = Xb.astype(np.float32)
    Xc, yc = make_classification(n_samples=200, n_features=6, random_state=0)
    empty = DataFrame.from_numpy(Xb[:0])
    models = [
        KMeans(k=3, maxIter=5, seed=1).fit(DataFrame.from_numpy(Xb)),
        PCA(k=2).fit(DataFrame.from_numpy(Xb)),
        LinearRegression().fit(DataFrame.from_numpy(Xc, yc.astype(np.float64))),
        LogisticRegression(maxIter=20).fit(
            DataFrame.from_numpy(Xc.astype(np.float32), yc.astype(np.float64))
        ),
        RandomForestClassifier(numTrees=3, maxDepth=3).fit(
            DataFrame.from_numpy(Xc.astype(np.float32), yc.astype(np.float64))
        ),
        UMAP(n_neighbors=10, n_epochs=20, random_state=1).fit(DataFrame.from_numpy(Xb)),
    ]
    for m in models:
        assert m.transform(empty).num_rows == 0


def _dist_variable_list_parquet(path: str):
    """Rank 1 reads zero row groups of a variable-width list<double> parquet
    (Spark writes array columns as variable lists): the empty shard reads as
    (0,0) and fit must still align its collective buffers (round-1 advisor
    finding: the 0-width shard deadlocked the fused all-reduce)."""
    from spark_rapids_ml_amd.data import DataFrame as ADF

    df = ADF.read_parquet(path, vector_cols=["features"])
    m = KMeans(k=3, maxIter=5, seed=1, featuresCol="features").fit(df)
    return m.cluster_centers_


def test_variable_list_parquet_empty_shard(tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    rng = np.random.default_rng(0)
    X = rng.normal(size=(64, 6))
    # variable-width list column (NOT fixed_size_list), single row group →
    # rank 1 gets the schema-preserving empty slice
    feats = pa.array([row.tolist() for row in X], type=pa.list_(pa.float64()))
    table = pa.table({"features": feats})
    p = str(tmp_path / "var_list.parquet")
    pq.write_table(table, p, row_group_size=len(X))
    results = run_distributed(_dist_variable_list_parquet, world_size=2, args=(p,))
    np.testing.assert_allclose(results[0], results[1])
    assert np.asarray(results[0]).shape[1] == 6
