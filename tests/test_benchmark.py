"""Benchmark harness + data-gen tests (reference benchmark/test_gen_data.py)."""

import json
import subprocess
import sys
import os

import numpy as np
import pytest

from benchmark import gen_data
from benchmark.benches import BENCHMARKS


def test_gen_blobs_stats():
    X, y = gen_data.gen_blobs(5000, 16, centers=4, cluster_std=0.5, seed=1)
    assert X.shape == (5000, 16)
    assert X.dtype == np.float32
    assert len(np.unique(y)) == 4


def test_gen_regression_recoverable():
    X, y = gen_data.gen_regression(2000, 20, noise=0.01, seed=2)
    w, *_ = np.linalg.lstsq(
        np.column_stack([X, np.ones(len(X))]), y, rcond=None
    )
    pred = np.column_stack([X, np.ones(len(X))]) @ w
    assert 1 - ((pred - y) ** 2).sum() / ((y - y.mean()) ** 2).sum() > 0.999


def test_gen_classification_separable():
    X, y = gen_data.gen_classification(2000, 10, n_classes=3, seed=3)
    assert set(np.unique(y)) == {0.0, 1.0, 2.0}


def test_gen_sparse_density():
    X, y = gen_data.gen_sparse_regression(1000, 50, density=0.1, seed=4)
    assert 0.05 < X.nnz / (1000 * 50) < 0.15


def test_gen_low_rank():
    X = gen_data.gen_low_rank_matrix(500, 32, effective_rank=4, seed=5)
    s = np.linalg.svd(X, compute_uv=False)
    assert s[0] / s[-1] > 10  # strongly low-rank spectrum


@pytest.mark.parametrize(
    "name",
    [
        "kmeans", "pca", "linear_regression", "logistic_regression",
        "random_forest_classifier", "random_forest_regressor",
        "nearest_neighbors", "approximate_nearest_neighbors",
        "dbscan", "umap",
    ],
)
def test_bench_smoke(name):
    argv = ["--num_rows", "2000", "--num_cols", "16"]
    if name == "kmeans":
        argv += ["--k", "8", "--maxIter", "3"]
    if name == "umap":
        argv += ["--n_epochs", "20"]
    if name.startswith("random_forest"):
        argv += ["--numTrees", "3", "--maxDepth", "3"]
    report = BENCHMARKS[name].run(argv)
    assert report["fit_sec"] > 0
    assert report["num_rows_total"] == 2000


def test_gen_data_cli(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "benchmark.gen_data", "blobs", "--num_rows", "500",
         "--num_cols", "8", "--output_dir", str(tmp_path / "blobs")],
        capture_output=True, text=True, cwd=repo, timeout=300,
    )
    assert out.returncode == 0, out.stderr
    from spark_rapids_ml_amd.data import DataFrame

    df = DataFrame.read_parquet(str(tmp_path / "blobs"), vector_cols=["features"])
    assert df.num_rows == 500


def test_gen_classification_redundant_and_clusters():
    X, y = gen_data.gen_classification(
        3000, 20, n_classes=2, n_informative=6, n_redundant=4,
        n_clusters_per_class=2, seed=7,
    )
    assert X.shape == (3000, 20)
    # redundant block must be an exact linear function of the informative one
    A, res, *_ = np.linalg.lstsq(X[:, :6], X[:, 6:10], rcond=None)
    recon = X[:, :6] @ A
    assert np.allclose(recon, X[:, 6:10], atol=1e-4)
    # remaining features are (mostly) independent noise
    B, *_ = np.linalg.lstsq(X[:, :6], X[:, 10:], rcond=None)
    resid = X[:, 10:] - X[:, :6] @ B
    assert resid.std() > 0.8


def test_gen_regression_effective_rank_and_coef():
    X, y, w = gen_data.gen_regression(
        2000, 24, n_informative=6, noise=0.0, effective_rank=4, seed=9,
        return_coef=True,
    )
    # low-rank spectrum: top-4 singular values carry most of the energy
    s = np.linalg.svd(np.asarray(X, np.float64), compute_uv=False)
    assert s[:4].sum() / s.sum() > 0.5
    # exact relation with zero noise
    np.testing.assert_allclose(X @ w + 0.5, y, rtol=1e-4, atol=1e-4)


def _dist_gen_union(kind):
    from benchmark import gen_data

    fn = getattr(gen_data, f"gen_{kind}")
    X, y = fn(4000, 12, seed=21)
    return X.sum(axis=0), X.shape[0]


def test_generators_shard_across_ranks():
    """2-rank generation must cover the global row count with disjoint,
    statistically consistent shards (reference gen_data_distributed's
    mapInPandas sharding)."""
    from .dist_utils import run_distributed

    results = run_distributed(_dist_gen_union, world_size=2, args=("classification",))
    assert sum(r[1] for r in results) == 4000
    assert all(r[1] > 0 for r in results)
