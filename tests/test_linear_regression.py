"""LinearRegression vs sklearn / closed forms (pattern: reference
tests/test_linear_model.py)."""

import numpy as np
import pytest
from sklearn.linear_model import ElasticNet, LinearRegression as SkOLS, Ridge

from spark_rapids_ml_amd import LinearRegression, LinearRegressionModel
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _data(n=400, d=10, seed=0, noise=0.1):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, d)) * rng.uniform(0.5, 2.0, size=d)
    w = rng.normal(size=d)
    y = X @ w + 3.0 + noise * rng.normal(size=n)
    return X.astype(np.float64), y.astype(np.float64), w


def test_ols_matches_sklearn():
    X, y, _ = _data()
    df = DataFrame.from_numpy(X, y)
    model = LinearRegression(regParam=0.0).fit(df)
    sk = SkOLS().fit(X, y)
    assert np.allclose(model.coefficients, sk.coef_, atol=1e-6)
    assert np.isclose(model.intercept, sk.intercept_, atol=1e-6)


def test_ols_no_intercept():
    X, y, _ = _data()
    model = LinearRegression(regParam=0.0, fitIntercept=False).fit(DataFrame.from_numpy(X, y))
    sk = SkOLS(fit_intercept=False).fit(X, y)
    assert np.allclose(model.coefficients, sk.coef_, atol=1e-6)
    assert model.intercept == 0.0


def test_ridge_matches_glmnet_objective():
    """Our ridge solves min 1/(2n)||ỹ-X̃w̃||² + λ/2||w̃||² in standardized
    space (Spark semantics). Verify against the closed form computed in
    numpy directly."""
    X, y, _ = _data(n=300, d=6, seed=1)
    lam = 0.3
    model = LinearRegression(regParam=lam, elasticNetParam=0.0).fit(
        DataFrame.from_numpy(X, y)
    )
    n = len(y)
    xbar, ybar = X.mean(0), y.mean()
    sx = X.std(0, ddof=1)
    sy = y.std(ddof=1)
    Xs = (X - xbar) / sx
    ys = (y - ybar) / sy
    wt = np.linalg.solve(Xs.T @ Xs / n + lam * np.eye(X.shape[1]), Xs.T @ ys / n)
    w = wt * sy / sx
    b = ybar - w @ xbar
    assert np.allclose(model.coefficients, w, atol=1e-6)
    assert np.isclose(model.intercept, b, atol=1e-6)


def test_elasticnet_matches_closed_objective():
    """CD result must minimize the stated objective: compare objective value
    against scikit-learn ElasticNet on the standardized problem."""
    X, y, _ = _data(n=300, d=8, seed=2)
    lam, alpha = 0.1, 0.5
    model = LinearRegression(
        regParam=lam, elasticNetParam=alpha, maxIter=500, tol=1e-9
    ).fit(DataFrame.from_numpy(X, y))
    n = len(y)
    xbar, ybar = X.mean(0), y.mean()
    sx = X.std(0, ddof=1)
    sy = y.std(ddof=1)
    Xs = (X - xbar) / sx
    ys = (y - ybar) / sy
    # sklearn ElasticNet: 1/(2n)||y-Xw||² + a*l1||w||₁ + a(1-l1)/2||w||²
    sk = ElasticNet(alpha=lam, l1_ratio=alpha, fit_intercept=False, tol=1e-10, max_iter=100000).fit(
        Xs, ys
    )
    w_expect = sk.coef_ * sy / sx
    assert np.allclose(model.coefficients, w_expect, atol=1e-4)


def test_sparsity_under_l1():
    X, y, _ = _data(n=200, d=20, seed=3)
    model = LinearRegression(regParam=0.5, elasticNetParam=1.0, maxIter=300).fit(
        DataFrame.from_numpy(X, y)
    )
    assert (np.abs(model.coefficients) < 1e-10).sum() > 0


def test_transform_prediction():
    X, y, _ = _data()
    model = LinearRegression().fit(DataFrame.from_numpy(X, y))
    out = model.transform(DataFrame.from_numpy(X))
    pred = np.asarray(out["prediction"])
    expect = X @ model.coefficients + model.intercept
    assert np.allclose(pred, expect, atol=1e-3)
    # r2 close to 1 on low-noise data
    ss_res = ((pred - y) ** 2).sum()
    ss_tot = ((y - y.mean()) ** 2).sum()
    assert 1 - ss_res / ss_tot > 0.98


def test_persistence(tmp_model_path):
    X, y, _ = _data(n=100)
    model = LinearRegression(regParam=0.1).fit(DataFrame.from_numpy(X, y))
    model.save(tmp_model_path)
    loaded = LinearRegressionModel.load(tmp_model_path)
    assert np.allclose(loaded.coefficients, model.coefficients)
    assert np.isclose(loaded.intercept, model.intercept)


def test_fit_multiple_single_pass():
    X, y, _ = _data(n=200, d=5)
    df = DataFrame.from_numpy(X, y)
    est = LinearRegression()
    maps = [
        {est.getParam("regParam"): 0.0},
        {est.getParam("regParam"): 0.5},
    ]
    models = dict(est.fitMultiple(df, maps))
    assert len(models) == 2
    assert not np.allclose(models[0].coefficients, models[1].coefficients)


def _dist_ols(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X, y, _ = _data(n=400, seed=seed)
    sl = slice(comm.rank, None, comm.world_size)
    model = LinearRegression().fit(DataFrame.from_numpy(X[sl], y[sl]))
    return np.asarray(model.coefficients), model.intercept


def test_ols_distributed_matches_single():
    results = run_distributed(_dist_ols, world_size=2, args=(0,))
    X, y, _ = _data(n=400, seed=0)
    sk = SkOLS().fit(X, y)
    for coef, icpt in results:
        assert np.allclose(coef, sk.coef_, atol=1e-6)
        assert np.isclose(icpt, sk.intercept_, atol=1e-6)


def test_summary_r2_matches_sklearn():
    from sklearn.linear_model import LinearRegression as SkLR

    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 5))
    y = X @ rng.normal(size=5) + 1.0 + 0.3 * rng.normal(size=300)
    m = LinearRegression().fit(DataFrame.from_numpy(X, y))
    sk = SkLR().fit(X, y)
    # f32 ingest => ~1e-7 agreement with sklearn's f64 score
    assert abs(m.summary.r2 - sk.score(X, y)) < 1e-5


def test_degenerate_columns_and_labels():
    """Constant features (sigma=0) and constant labels stay finite through
    every solver path (standardization divides by sigma internally)."""
    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 5))
    X[:, 2] = 7.0
    y = X @ np.array([1.0, 2.0, 0.0, 3.0, -1.0]) + 0.5
    for kw in (dict(regParam=0.1), dict(regParam=0.1, elasticNetParam=0.5, maxIter=20)):
        m = LinearRegression(**kw).fit(DataFrame.from_numpy(X, y))
        assert np.isfinite(np.asarray(m.coefficients)).all()
        assert abs(np.asarray(m.coefficients)[2]) < 1e-6  # constant col gets 0
    m = LinearRegression(regParam=0.1, elasticNetParam=0.5, maxIter=10).fit(
        DataFrame.from_numpy(X, np.full(200, 3.0))
    )
    assert np.isfinite(np.asarray(m.coefficients)).all()
    assert abs(m.intercept - 3.0) < 1e-8


def test_ridge_standardized_penalty_objective():
    """standardization=True ridge: raw coefficients minimize
    1/(2n)||y-Xw-b||^2 + lam/2 * sum((w_j*sigma_j)^2) (the label-sigma
    factors cancel in raw space). Verified against a scipy optimum."""
    from scipy.optimize import minimize

    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 4)) * np.array([1.0, 4.0, 0.3, 2.0])
    y = X @ np.array([1.0, -0.5, 2.0, 0.7]) + 1.5 + 0.2 * rng.normal(size=300)
    lam = 0.3
    model = LinearRegression(regParam=lam).fit(DataFrame.from_numpy(X, y))
    sigma = X.std(axis=0, ddof=1)
    n = len(y)

    def obj(wb):
        w, b = wb[:4], wb[4]
        r = y - X @ w - b
        return float(r @ r / (2 * n) + lam / 2 * np.sum((w * sigma) ** 2))

    ours = obj(np.concatenate([np.asarray(model.coefficients), [model.intercept]]))
    ref = minimize(obj, np.zeros(5), method="L-BFGS-B", options={"maxiter": 2000}).fun
    assert ours <= ref * (1 + 1e-6), (ours, ref)


def test_elasticnet_matches_sklearn_on_standardized_data():
    """On unit-variance features the Spark elastic-net objective maps onto
    sklearn's with alpha' = lam*(a*sy + (1-a)), l1_ratio' = a*sy/(a*sy+(1-a))
    where sy is the label std (Spark standardizes the label; the L1 term
    scales linearly in sy, the L2/loss terms quadratically)."""
    from sklearn.linear_model import ElasticNet as SkEN

    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 6))
    X = (X - X.mean(0)) / X.std(0, ddof=1)  # unit sigma (ddof=1)
    y = X @ np.array([2.0, -1.0, 0.0, 0.0, 1.5, 0.0]) + 0.5 + 0.1 * rng.normal(size=400)
    lam, a = 0.1, 0.5
    model = LinearRegression(regParam=lam, elasticNetParam=a, maxIter=500, tol=1e-12).fit(
        DataFrame.from_numpy(X, y)
    )
    sy = y.std(ddof=1)
    alpha_sk = lam * (a * sy + (1 - a))
    l1r_sk = a * sy / (a * sy + (1 - a))
    sk = SkEN(alpha=alpha_sk, l1_ratio=l1r_sk, max_iter=50000, tol=1e-12).fit(X, y)
    assert np.allclose(np.asarray(model.coefficients), sk.coef_, atol=2e-4), (
        np.asarray(model.coefficients), sk.coef_)
    assert abs(model.intercept - sk.intercept_) < 1e-3
