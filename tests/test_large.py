"""Larger-scale tests (reference tests_large/: full-objective comparison of
logistic regression against the CPU implementation's objectiveHistory,
test_large_logistic_regression.py:39-60). Marked slow."""

import numpy as np
import pytest
from sklearn.datasets import make_classification

from spark_rapids_ml_amd import LogisticRegression
from spark_rapids_ml_amd.data import DataFrame


@pytest.mark.slow
def test_logreg_objective_history_reaches_sklearn():
    X, y = make_classification(
        n_samples=50000, n_features=128, n_informative=64, random_state=0
    )
    X = X.astype(np.float64)
    y = y.astype(np.float64)
    lam = 1e-5
    model = LogisticRegression(
        regParam=lam, standardization=False, maxIter=200, tol=1e-10
    ).fit(DataFrame.from_numpy(X, y))
    hist = model.objectiveHistory
    assert len(hist) >= 2
    assert hist[0] > hist[-1]  # monotone-ish descent
    assert np.all(np.diff(hist) <= 1e-12)

    from sklearn.linear_model import LogisticRegression as SkLR

    sk = SkLR(penalty="l2", C=1.0 / (lam * len(y)), max_iter=2000, tol=1e-10).fit(X, y)

    def obj(w, b):
        z = X @ w + b
        t = 2 * y - 1
        return float(np.mean(np.logaddexp(0, -t * z)) + lam / 2 * (w @ w))

    assert hist[-1] <= obj(sk.coef_[0], sk.intercept_[0]) * (1 + 1e-4)


def test_logreg_objective_history_small():
    X, y = make_classification(n_samples=2000, n_features=32, random_state=1)
    model = LogisticRegression(regParam=0.01, maxIter=50).fit(
        DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    )
    hist = model.objectiveHistory
    assert len(hist) == model._model_attributes["n_iter_"] + 1
    assert hist[-1] <= hist[0]


@pytest.mark.gpu
def test_large_logreg_objective_10m_gpu():
    """tests_large analog (reference test_large_logistic_regression.py:39-60):
    at 10M x 64 the GPU fit's final objective must descend monotonically and
    beat a subsample-estimated objective bound."""
    from benchmark import gen_data

    X, y = gen_data.gen_classification(
        10_000_000, 64, n_classes=2, n_informative=32, seed=0
    )
    lam = 1e-6
    model = LogisticRegression(
        regParam=lam, standardization=False, maxIter=60, tol=1e-9
    ).fit(DataFrame.from_numpy(X, y))
    hist = model.objectiveHistory
    assert len(hist) >= 3
    assert np.all(np.diff(hist) <= 1e-12)

    # objective sanity on a held subsample with the fitted coefficients
    w = np.asarray(model.coefficients, np.float64).ravel()
    b = float(np.asarray(model.intercept_).ravel()[0])
    sub = slice(0, 500_000)
    z = X[sub].astype(np.float64) @ w + b
    t = 2 * y[sub] - 1
    sub_obj = float(np.mean(np.logaddexp(0, -t * z)))
    assert abs(sub_obj - hist[-1]) < 0.02
    # accuracy must be far above chance on informative data
    acc = float(((z > 0) == (y[sub] > 0)).mean())
    assert acc > 0.8


@pytest.mark.gpu
def test_large_linreg_exact_recovery_20m_gpu():
    """20M x 32 linear regression recovers the planted coefficients."""
    from benchmark import gen_data

    from spark_rapids_ml_amd import LinearRegression

    X, y, w = gen_data.gen_regression(
        20_000_000, 32, n_informative=16, noise=0.5, seed=1, return_coef=True
    )
    m = LinearRegression().fit(DataFrame.from_numpy(X, y))
    got = np.asarray(m.coefficients, np.float64)
    np.testing.assert_allclose(got, w, atol=2e-3)
    assert np.isclose(m.intercept, 0.5, atol=2e-3)
