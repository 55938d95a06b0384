import numpy as np
import pytest

from spark_rapids_ml_amd.data import DataFrame, extract_features


def test_from_numpy_and_columns():
    X = np.random.rand(10, 4).astype(np.float32)
    y = np.arange(10.0)
    df = DataFrame.from_numpy(X, y)
    assert df.columns == ["features", "label"]
    assert df.num_rows == 10
    assert np.allclose(df["features"], X)


def test_extract_features_vector_col():
    X = np.random.rand(8, 3).astype(np.float64)
    df = DataFrame.from_numpy(X)
    out = extract_features(df, "features", None, float32_inputs=True)
    assert out.dtype == np.float32
    out64 = extract_features(df, "features", None, float32_inputs=False)
    assert out64.dtype == np.float64


def test_extract_features_multi_cols():
    df = DataFrame({"a": np.arange(5.0), "b": np.arange(5.0) * 2})
    out = extract_features(df, None, ["a", "b"], True)
    assert out.shape == (5, 2)
    assert np.allclose(out[:, 1], np.arange(5) * 2)


def test_with_column_select_drop():
    df = DataFrame({"a": np.arange(4)})
    df2 = df.with_column("b", np.ones(4))
    assert set(df2.columns) == {"a", "b"}
    assert df2.select("b").columns == ["b"]
    assert df2.drop("a").columns == ["b"]


def test_row_mismatch_raises():
    with pytest.raises(ValueError):
        DataFrame({"a": np.arange(3), "b": np.arange(4)})


def test_parquet_roundtrip(tmp_path):
    X = np.random.rand(20, 4).astype(np.float32)
    y = np.random.rand(20)
    df = DataFrame.from_numpy(X, y)
    path = str(tmp_path / "data")
    df.write_parquet(path)
    back = DataFrame.read_parquet(path, vector_cols=["features"])
    assert back.num_rows == 20
    assert np.allclose(np.asarray(back["features"]), X)
    assert np.allclose(np.asarray(back["label"]), y)


def test_to_pandas():
    X = np.random.rand(6, 2).astype(np.float32)
    df = DataFrame.from_numpy(X)
    pdf = df.to_pandas()
    assert len(pdf) == 6
    assert np.allclose(np.stack(pdf["features"].to_list()), X)


def _dist_parquet_shard(path):
    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    df = DataFrame.read_parquet(path, vector_cols=["features"])
    ids = np.asarray(df["id"]) if "id" in df.columns else None
    return comm.rank, len(df), ids


def test_read_parquet_shards_across_ranks(tmp_path):
    from tests.dist_utils import run_distributed

    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    X = rng.normal(size=(5000, 8)).astype(np.float32)
    ids = np.arange(5000, dtype=np.int64)
    full = DataFrame({"features": X, "id": ids})
    out = str(tmp_path / "shard_ds")
    # row_group_rows small enough to give every rank several groups
    full.write_parquet(out, row_group_rows=500)

    results = run_distributed(_dist_parquet_shard, world_size=2, args=(out,))
    counts = {rank: n for rank, n, _ in results}
    assert sum(counts.values()) == 5000
    assert min(counts.values()) > 0
    # shards are disjoint and together cover every row exactly once
    all_ids = np.concatenate([i for _, _, i in results])
    assert np.array_equal(np.sort(all_ids), ids)


def _dist_count(_):
    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    n_local = 100 + 50 * comm.rank  # uneven shards
    df = DataFrame({"x": np.arange(n_local, dtype=np.float64)})
    return df.num_rows, df.count()


def test_count_is_global(tmp_path):
    from tests.dist_utils import run_distributed

    results = run_distributed(_dist_count, world_size=2, args=(None,))
    assert results[0] == (100, 250)
    assert results[1] == (150, 250)


def test_sparse_input_densified_for_dense_estimators():
    """CSR features to estimators without a sparse path densify on ingest
    (Spark SparseVector.toArray behavior); LogisticRegression keeps CSR."""
    import scipy.sparse as sp

    from spark_rapids_ml_amd import KMeans, LinearRegression, PCA

    rng = np.random.default_rng(0)
    X = sp.random(150, 10, density=0.3, format="csr", dtype=np.float32, random_state=0)
    y = rng.normal(size=150)
    km = KMeans(k=3, maxIter=5, seed=1).fit(DataFrame.from_numpy(X))
    assert km.cluster_centers_.shape == (3, 10)
    assert PCA(k=2).fit(DataFrame.from_numpy(X)).components_.shape == (2, 10)
    lr = LinearRegression().fit(DataFrame.from_numpy(X, y))
    # same result as explicit densify
    lr_d = LinearRegression().fit(DataFrame.from_numpy(np.asarray(X.todense()), y))
    assert np.allclose(lr.coefficients, lr_d.coefficients, atol=1e-6)
