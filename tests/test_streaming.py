"""Capacity path: fits larger than the device-data cap stream chunked
host->device and must match the in-memory fit (reference reserved-GPU-memory
streaming ingest, utils.py:403-522; conf gpu_mem_ratio_for_data).

SRML_STREAM_CAP_BYTES forces an absolute cap so the path is exercised on CPU
(the ratio form needs a CUDA device total)."""

import numpy as np
import pytest
from sklearn.datasets import make_classification, make_regression

from spark_rapids_ml_amd import LinearRegression, LogisticRegression, PCA
from spark_rapids_ml_amd.data import DataFrame


@pytest.fixture()
def small_cap(monkeypatch):
    # dataset below is ~16x this cap -> many chunks
    monkeypatch.setenv("SRML_STREAM_CAP_BYTES", str(64 * 1024))
    yield


def test_streamed_moments_match_dense():
    import torch

    from spark_rapids_ml_amd.streaming import streamed_moments

    rng = np.random.default_rng(0)
    X = rng.normal(size=(3000, 12)).astype(np.float32)
    y = rng.normal(size=3000).astype(np.float32)
    mom = streamed_moments(X, y, torch.device("cpu"), cap_bytes=32 * 1024)
    Xt = torch.from_numpy(X).to(torch.float64)
    yt = torch.from_numpy(y).to(torch.float64)
    # per-chunk f32 gram accumulated in f64 (same precision class as the
    # in-memory f32 MFMA gram): near-zero cross terms need an absolute floor
    np.testing.assert_allclose(mom["G"].numpy(), (Xt.T @ Xt).numpy(), rtol=1e-4, atol=5e-2)
    np.testing.assert_allclose(mom["Xty"].numpy(), (Xt.T @ yt).numpy(), rtol=1e-4, atol=5e-2)
    np.testing.assert_allclose(mom["xsum"].numpy(), Xt.sum(0).numpy(), rtol=1e-5, atol=1e-3)
    assert np.isclose(float(mom["ysum"]), float(yt.sum()), rtol=1e-5, atol=1e-3)


def test_pca_streaming_matches_inmemory(small_cap):
    rng = np.random.default_rng(0)
    X = rng.normal(size=(4000, 16)).astype(np.float32)  # 256 KB = 4x cap
    df = DataFrame.from_numpy(X)
    m_stream = PCA(k=4).fit(df)

    import os

    del os.environ["SRML_STREAM_CAP_BYTES"]
    m_mem = PCA(k=4).fit(df)
    np.testing.assert_allclose(
        np.abs(m_stream.components_), np.abs(m_mem.components_), rtol=1e-4, atol=1e-6
    )
    np.testing.assert_allclose(
        m_stream.explained_variance_, m_mem.explained_variance_, rtol=1e-4
    )


def test_linreg_streaming_matches_inmemory(small_cap):
    X, y = make_regression(n_samples=4000, n_features=16, noise=2.0, random_state=0)
    X = X.astype(np.float32)
    df = DataFrame.from_numpy(X, y)
    m_stream = LinearRegression().fit(df)

    import os

    del os.environ["SRML_STREAM_CAP_BYTES"]
    m_mem = LinearRegression().fit(df)
    np.testing.assert_allclose(
        np.asarray(m_stream.coefficients), np.asarray(m_mem.coefficients),
        rtol=1e-4, atol=1e-5,
    )
    assert np.isclose(m_stream.intercept, m_mem.intercept, rtol=1e-4, atol=1e-5)


def test_logreg_streaming_matches_inmemory(small_cap):
    X, y = make_classification(
        n_samples=4000, n_features=16, n_informative=8, random_state=0
    )
    df = DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    m_stream = LogisticRegression(maxIter=40, regParam=0.01).fit(df)

    import os

    del os.environ["SRML_STREAM_CAP_BYTES"]
    m_mem = LogisticRegression(maxIter=40, regParam=0.01).fit(df)
    np.testing.assert_allclose(
        np.asarray(m_stream.coefficients).ravel(),
        np.asarray(m_mem.coefficients).ravel(),
        rtol=5e-3,
        atol=1e-4,
    )
    pred_s = np.asarray(m_stream.transform(df)[m_stream.getOrDefault("predictionCol")])
    pred_m = np.asarray(m_mem.transform(df)[m_mem.getOrDefault("predictionCol")])
    assert (pred_s == pred_m).mean() > 0.999


def test_ridge_elasticnet_streaming(small_cap):
    X, y = make_regression(n_samples=4000, n_features=16, noise=2.0, random_state=1)
    X = X.astype(np.float32)
    df = DataFrame.from_numpy(X, y)
    m_ridge = LinearRegression(regParam=0.5).fit(df)
    m_enet = LinearRegression(regParam=0.1, elasticNetParam=0.5, maxIter=50).fit(df)

    import os

    del os.environ["SRML_STREAM_CAP_BYTES"]
    r2 = LinearRegression(regParam=0.5).fit(df)
    e2 = LinearRegression(regParam=0.1, elasticNetParam=0.5, maxIter=50).fit(df)
    np.testing.assert_allclose(
        np.asarray(m_ridge.coefficients), np.asarray(r2.coefficients), rtol=1e-4
    )
    np.testing.assert_allclose(
        np.asarray(m_enet.coefficients), np.asarray(e2.coefficients), rtol=1e-3, atol=1e-5
    )
