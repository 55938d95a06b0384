"""PCA vs sklearn (pattern: reference tests/test_pca.py)."""

import numpy as np
import pytest
from sklearn.datasets import make_blobs
from sklearn.decomposition import PCA as SkPCA

from spark_rapids_ml_amd import PCA, PCAModel
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _data(n=500, d=16, seed=0):
    rng = np.random.default_rng(seed)
    W = rng.normal(size=(4, d))
    Z = rng.normal(size=(n, 4))
    X = (Z @ W + 0.05 * rng.normal(size=(n, d)) + rng.normal(size=d)).astype(np.float64)
    return X


def _align_signs(A, B):
    """Flip rows of A to best match B (component sign is conventional)."""
    out = A.copy()
    for i in range(A.shape[0]):
        if np.dot(A[i], B[i]) < 0:
            out[i] = -A[i]
    return out


def test_pca_components_match_sklearn():
    X = _data()
    model = PCA(k=3).fit(DataFrame.from_numpy(X))
    sk = SkPCA(n_components=3).fit(X)
    ours = _align_signs(np.asarray(model.components_), sk.components_)
    assert np.allclose(ours, sk.components_, atol=1e-5)
    assert np.allclose(model._model_attributes["explained_variance_"], sk.explained_variance_, rtol=1e-6)
    assert np.allclose(
        model._model_attributes["explained_variance_ratio_"],
        sk.explained_variance_ratio_,
        rtol=1e-6,
    )
    assert np.allclose(model.mean_, X.mean(axis=0), atol=1e-8)


def test_pca_transform_spark_semantics():
    # Spark projects WITHOUT centering (reference feature.py:438-449)
    X = _data(n=100)
    model = PCA(k=2).fit(DataFrame.from_numpy(X))
    out = model.transform(DataFrame.from_numpy(X.astype(np.float32)))
    proj = np.asarray(out["pca_features"])
    expect = X.astype(np.float32) @ np.asarray(model.components_).T.astype(np.float32)
    assert np.allclose(proj, expect, atol=1e-3)


def test_pca_sign_convention():
    # largest-|v| element of every component is positive
    X = _data()
    model = PCA(k=3).fit(DataFrame.from_numpy(X))
    comp = np.asarray(model.components_)
    for row in comp:
        assert row[np.abs(row).argmax()] > 0


def test_pca_persistence(tmp_model_path):
    X = _data(n=100)
    model = PCA(k=2).fit(DataFrame.from_numpy(X))
    model.save(tmp_model_path)
    loaded = PCAModel.load(tmp_model_path)
    assert np.allclose(loaded.components_, model.components_)
    assert loaded.getOrDefault("k") == 2


def test_pca_k_too_large_raises():
    X = _data(n=10, d=4)
    with pytest.raises(ValueError):
        PCA(k=8).fit(DataFrame.from_numpy(X))


def _dist_pca_fit(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X = _data(n=500, seed=seed)
    shard = X[comm.rank :: comm.world_size]
    model = PCA(k=3).fit(DataFrame.from_numpy(shard))
    return np.asarray(model.components_), np.asarray(model.mean_)


def test_pca_distributed_matches_single():
    results = run_distributed(_dist_pca_fit, world_size=2, args=(0,))
    comp0, mean0 = results[0]
    comp1, _ = results[1]
    assert np.allclose(comp0, comp1)
    X = _data(n=500, seed=0)
    single = PCA(k=3).fit(DataFrame.from_numpy(X))
    assert np.allclose(comp0, np.asarray(single.components_), atol=1e-6)
    assert np.allclose(mean0, X.mean(axis=0), atol=1e-8)


def test_pca_input_col_api():
    X = _data(n=100)
    df = DataFrame({"feat_vec": X})
    model = PCA(k=2).setInputCol("feat_vec").setOutputCol("proj").fit(df)
    out = model.transform(df)
    assert np.asarray(out["proj"]).shape == (100, 2)


def test_pca_rectangle_fewer_rows_than_cols():
    """n < d (reference test_fit_rectangle): covariance is rank-deficient but
    the top components still match sklearn."""
    from sklearn.decomposition import PCA as SkPCA

    rng = np.random.default_rng(0)
    X = (rng.normal(size=(60, 200)) @ np.diag(np.linspace(3, 0.1, 200))).astype(np.float32)
    model = PCA(k=4).fit(DataFrame.from_numpy(X))
    sk = SkPCA(n_components=4).fit(X)
    for i in range(4):
        dot = abs(np.dot(model.components_[i], sk.components_[i]))
        assert dot > 0.99, (i, dot)
    assert np.allclose(
        model.explained_variance_ratio_, sk.explained_variance_ratio_, atol=1e-4
    )
