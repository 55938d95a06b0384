"""CrossValidator + evaluators (pattern: reference tests/test_tuning.py)."""

import numpy as np
import pytest
from sklearn.datasets import make_classification, make_regression

from spark_rapids_ml_amd import LinearRegression, LogisticRegression
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.evaluation import (
    BinaryClassificationEvaluator,
    MulticlassClassificationEvaluator,
    RegressionEvaluator,
)
from spark_rapids_ml_amd.tuning import CrossValidator, CrossValidatorModel, ParamGridBuilder


def test_param_grid_builder():
    lr = LinearRegression()
    grid = (
        ParamGridBuilder()
        .addGrid(lr.getParam("regParam"), [0.0, 0.1])
        .addGrid(lr.getParam("elasticNetParam"), [0.0, 0.5])
        .build()
    )
    assert len(grid) == 4


def test_cv_regression_selects_best():
    X, y = make_regression(n_samples=400, n_features=8, noise=1.0, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    lr = LinearRegression()
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 100.0]).build()
    cv = CrossValidator(
        estimator=lr,
        estimatorParamMaps=grid,
        evaluator=RegressionEvaluator(metricName="rmse"),
        numFolds=3,
        seed=7,
    )
    model = cv.fit(df)
    assert len(model.avgMetrics) == 2
    # unregularized fits clean linear data far better than λ=100
    assert model.avgMetrics[0] < model.avgMetrics[1]
    assert model.bestModel.getOrDefault("regParam") == 0.0
    out = model.transform(df)
    assert "prediction" in out.columns


def test_cv_classification():
    X, y = make_classification(n_samples=400, n_features=8, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    lr = LogisticRegression(maxIter=50)
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.01, 10.0]).build()
    cv = CrossValidator(
        estimator=lr,
        estimatorParamMaps=grid,
        evaluator=MulticlassClassificationEvaluator(metricName="f1"),
        numFolds=3,
    )
    model = cv.fit(df)
    assert model.bestModel.getOrDefault("regParam") == 0.01


def test_evaluators_basic():
    rng = np.random.default_rng(0)
    y = rng.integers(0, 2, 200).astype(np.float64)
    scores = np.where(y == 1, 0.8, 0.2) + 0.1 * rng.random(200)
    df = DataFrame(
        {
            "label": y,
            "prediction": (scores > 0.5).astype(np.float64),
            "rawPrediction": np.stack([-scores, scores], axis=1),
        }
    )
    auc = BinaryClassificationEvaluator().evaluate(df)
    assert auc > 0.9
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(df)
    assert acc > 0.9


def test_cv_model_persistence(tmp_path):
    X, y = make_regression(n_samples=200, n_features=6, noise=1.0, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    lr = LinearRegression()
    cv = CrossValidator(
        estimator=lr,
        estimatorParamMaps=ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 1.0]).build(),
        evaluator=RegressionEvaluator(metricName="rmse"),
        numFolds=2,
    )
    model = cv.fit(df)
    path = str(tmp_path / "cv")
    model.save(path)
    loaded = CrossValidatorModel.load(path)
    assert loaded.avgMetrics == model.avgMetrics
    assert np.allclose(loaded.bestModel.coefficients, model.bestModel.coefficients)


def _dist_cv(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X, y = make_regression(n_samples=300, n_features=6, noise=1.0, random_state=seed)
    sl = slice(comm.rank, None, comm.world_size)
    df = DataFrame.from_numpy(X[sl].astype(np.float64), y[sl].astype(np.float64))
    lr = LinearRegression()
    cv = CrossValidator(
        estimator=lr,
        estimatorParamMaps=ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 50.0]).build(),
        evaluator=RegressionEvaluator(metricName="rmse"),
        numFolds=2,
    )
    model = cv.fit(df)
    return model.avgMetrics, float(model.bestModel.getOrDefault("regParam"))


def test_cv_distributed():
    from .dist_utils import run_distributed

    results = run_distributed(_dist_cv, world_size=2, args=(0,))
    for metrics, best in results:
        assert len(metrics) == 2
        assert best == 0.0
    # ranks agree on metrics (collectives aligned)
    assert np.allclose(results[0][0], results[1][0])


def test_cross_validator_estimator_persistence(tmp_model_path):
    import os

    lr = LinearRegression()
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 0.1]).build()
    cv = CrossValidator(lr, grid, RegressionEvaluator(metricName="mae"), numFolds=2, seed=7)
    cv.save(os.path.join(tmp_model_path, "cv"))
    loaded = CrossValidator.load(os.path.join(tmp_model_path, "cv"))
    assert loaded.getOrDefault("numFolds") == 2
    assert loaded.getEvaluator().getOrDefault("metricName") == "mae"
    maps = loaded.getEstimatorParamMaps()
    assert len(maps) == 2
    assert sorted(v for pm in maps for v in pm.values()) == [0.0, 0.1]


def test_cv_smaller_is_better_metric():
    """CV must pick argmin for loss-like metrics (isLargerBetter False)."""
    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 5))
    y = X @ rng.normal(size=5) + 0.05 * rng.normal(size=400)
    df = DataFrame.from_numpy(X, y)
    lr = LinearRegression()
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 10.0]).build()
    ev = RegressionEvaluator(metricName="rmse")
    assert not ev.isLargerBetter()
    cvm = CrossValidator(lr, grid, ev, numFolds=2, seed=1).fit(df)
    # heavy ridge on well-posed data must lose; best = regParam 0.0
    assert cvm.bestModel.getRegParam() == 0.0
    assert cvm.avgMetrics[0] < cvm.avgMetrics[1]


def test_cv_fold_col():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 4))
    y = X @ rng.normal(size=4)
    folds = np.arange(300) % 3
    df = DataFrame({"features": X, "label": y, "fold": folds.astype(np.int64)})
    lr = LinearRegression()
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 0.1]).build()
    cvm = CrossValidator(lr, grid, RegressionEvaluator(), numFolds=3, foldCol="fold").fit(df)
    assert len(cvm.avgMetrics) == 2
    bad = DataFrame({"features": X, "label": y, "fold": (folds + 5).astype(np.int64)})
    with pytest.raises(ValueError, match="foldCol"):
        CrossValidator(lr, grid, RegressionEvaluator(), numFolds=3, foldCol="fold").fit(bad)


def test_transform_evaluate_single_pass_counts_extractions(monkeypatch):
    """Combined-model _transformEvaluate must extract features ONCE for all
    models (reference one-job multi-model evaluate, core.py:1572-1693)."""
    import spark_rapids_ml_amd.core as core_mod
    from spark_rapids_ml_amd import LogisticRegression
    from spark_rapids_ml_amd.core import Model
    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator

    X, y = make_classification(n_samples=300, n_features=6, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float64), y.astype(np.float64))
    est = LogisticRegression(maxIter=30)
    maps = [{est.regParam: 0.0}, {est.regParam: 1.0}, {est.regParam: 10.0}]
    models = [m for _, m in est.fitMultiple(df, maps)]

    calls = {"n": 0}
    orig = core_mod.extract_features

    def counting(*a, **k):
        calls["n"] += 1
        return orig(*a, **k)

    monkeypatch.setattr(core_mod, "extract_features", counting)
    combined = Model._combine(models)
    ev = MulticlassClassificationEvaluator(metricName="accuracy")
    ms = combined._transformEvaluate(df, ev)
    assert len(ms) == 3
    assert calls["n"] == 1, f"expected 1 feature extraction, saw {calls['n']}"
    # heavier regularization should not beat the unregularized fit here
    assert ms[0] >= ms[2] - 1e-9


def test_fit_multiple_grids_vary_for_all_estimators():
    """Grid values must actually reach every estimator's fit (the round-1
    advisor found logreg ignoring its grid; pin the whole family)."""
    from spark_rapids_ml_amd import KMeans, RandomForestClassifier

    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 8)).astype(np.float32)
    df = DataFrame.from_numpy(X)
    km = KMeans(maxIter=5, seed=1)
    models = [m for _, m in km.fitMultiple(df, [{km.k: 2}, {km.k: 5}])]
    assert models[0].cluster_centers_.shape[0] == 2
    assert models[1].cluster_centers_.shape[0] == 5

    y = (X[:, 0] > 0).astype(np.float64)
    dfc = DataFrame.from_numpy(X, y)
    rf = RandomForestClassifier(maxDepth=3, seed=1)
    models = [m for _, m in rf.fitMultiple(dfc, [{rf.numTrees: 3}, {rf.numTrees: 7}])]
    assert models[0].numTrees == 3
    assert models[1].numTrees == 7

    lr_X, lr_y = X.astype(np.float64), X[:, 0] * 2.0 + 1.0
    from spark_rapids_ml_amd import LinearRegression

    lr = LinearRegression(maxIter=20)
    models = [
        m for _, m in lr.fitMultiple(
            DataFrame.from_numpy(lr_X, lr_y),
            [{lr.regParam: 0.0}, {lr.regParam: 100.0}],
        )
    ]
    c0 = np.linalg.norm(np.asarray(models[0].coefficients))
    c1 = np.linalg.norm(np.asarray(models[1].coefficients))
    assert c1 < c0  # heavy ridge shrinks
