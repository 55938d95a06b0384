"""RandomForest classifier/regressor (pattern: reference
tests/test_random_forest.py — accuracy/r2 parity vs sklearn)."""

import numpy as np
import pytest
from sklearn.datasets import make_classification, make_regression
from sklearn.ensemble import (
    RandomForestClassifier as SkRFC,
    RandomForestRegressor as SkRFR,
)

from spark_rapids_ml_amd import (
    RandomForestClassifier,
    RandomForestClassificationModel,
    RandomForestRegressor,
    RandomForestRegressionModel,
)
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _cls_data(n=600, d=10, seed=0):
    X, y = make_classification(
        n_samples=n, n_features=d, n_informative=6, random_state=seed
    )
    return X.astype(np.float32), y.astype(np.float64)


def _reg_data(n=600, d=10, seed=0):
    X, y = make_regression(n_samples=n, n_features=d, n_informative=6, noise=5.0, random_state=seed)
    return X.astype(np.float32), y.astype(np.float64)


def test_rfc_accuracy_close_to_sklearn():
    X, y = _cls_data()
    model = RandomForestClassifier(numTrees=30, maxDepth=8, maxBins=64, seed=0).fit(
        DataFrame.from_numpy(X, y)
    )
    out = model.transform(DataFrame.from_numpy(X))
    acc = (np.asarray(out["prediction"]) == y).mean()
    sk = SkRFC(n_estimators=30, max_depth=8, random_state=0).fit(X, y)
    sk_acc = (sk.predict(X) == y).mean()
    assert acc > sk_acc - 0.05, f"ours {acc} vs sklearn {sk_acc}"
    probs = np.asarray(out["probability"])
    assert probs.shape == (len(y), 2)
    assert np.allclose(probs.sum(axis=1), 1.0, atol=1e-5)


def test_rfr_r2_close_to_sklearn():
    X, y = _reg_data()
    model = RandomForestRegressor(numTrees=30, maxDepth=8, maxBins=64, seed=0).fit(
        DataFrame.from_numpy(X, y)
    )
    out = model.transform(DataFrame.from_numpy(X))
    pred = np.asarray(out["prediction"])
    r2 = 1 - ((pred - y) ** 2).sum() / ((y - y.mean()) ** 2).sum()
    sk = SkRFR(n_estimators=30, max_depth=8, random_state=0).fit(X, y)
    sk_r2 = sk.score(X, y)
    assert r2 > sk_r2 - 0.1, f"ours {r2} vs sklearn {sk_r2}"


def test_rfc_multiclass():
    X, y = make_classification(
        n_samples=500,
        n_features=8,
        n_informative=6,
        n_classes=3,
        n_clusters_per_class=1,
        random_state=1,
    )
    X = X.astype(np.float32)
    y = y.astype(np.float64)
    model = RandomForestClassifier(numTrees=20, maxDepth=8, seed=1).fit(
        DataFrame.from_numpy(X, y)
    )
    assert model.numClasses == 3
    out = model.transform(DataFrame.from_numpy(X))
    acc = (np.asarray(out["prediction"]) == y).mean()
    assert acc > 0.85


def test_rfc_bad_labels_raise():
    X = np.random.rand(50, 4).astype(np.float32)
    y = np.array([0.5] * 50)
    with pytest.raises(ValueError):
        RandomForestClassifier(numTrees=2).fit(DataFrame.from_numpy(X, y))


def test_rf_num_trees_and_depth():
    X, y = _cls_data(n=200)
    model = RandomForestClassifier(numTrees=7, maxDepth=3, seed=0).fit(
        DataFrame.from_numpy(X, y)
    )
    assert model.numTrees == 7
    assert model.numFeatures == 10
    for t in model.trees:
        assert t["feature"].shape == t["left"].shape
        # depth-3 tree has at most 2^4-1 nodes
        assert t["feature"].shape[0] <= 15


def test_rf_persistence(tmp_model_path):
    X, y = _cls_data(n=200)
    model = RandomForestClassifier(numTrees=5, maxDepth=4, seed=0).fit(
        DataFrame.from_numpy(X, y)
    )
    model.save(tmp_model_path)
    loaded = RandomForestClassificationModel.load(tmp_model_path)
    assert loaded.numTrees == 5
    o1 = model.transform(DataFrame.from_numpy(X))
    o2 = loaded.transform(DataFrame.from_numpy(X))
    assert np.array_equal(np.asarray(o1["prediction"]), np.asarray(o2["prediction"]))


def test_rf_dump_as_json():
    X, y = _cls_data(n=100)
    model = RandomForestClassifier(numTrees=2, maxDepth=3, seed=0).fit(
        DataFrame.from_numpy(X, y)
    )
    import json

    dumped = json.loads(model.dump_as_json())
    assert len(dumped) == 2
    assert "feature" in dumped[0]


def _dist_rf(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X, y = _cls_data(n=600, seed=seed)
    sl = slice(comm.rank, None, comm.world_size)
    model = RandomForestClassifier(numTrees=10, maxDepth=6, seed=0).fit(
        DataFrame.from_numpy(X[sl], y[sl])
    )
    out = model.transform(DataFrame.from_numpy(X, y))
    return model.numTrees, (np.asarray(out["prediction"]) == y).mean()


def test_rf_distributed_trees_split_across_ranks():
    results = run_distributed(_dist_rf, world_size=2, args=(0,))
    for n_trees, acc in results:
        assert n_trees == 10  # merged forest has all trees
        assert acc > 0.85


def test_rf_feature_importances_and_debug_string():
    X, y = _cls_data(n=400, d=8)
    # feature 0 carries all signal
    X[:, 0] = y * 4 + np.random.default_rng(0).normal(0, 0.1, len(y))
    model = RandomForestClassifier(
        numTrees=10, maxDepth=4, seed=0, featureSubsetStrategy="all"
    ).fit(DataFrame.from_numpy(X, y))
    imp = model.featureImportances
    assert imp.shape == (8,)
    assert abs(imp.sum() - 1.0) < 1e-6
    assert imp.argmax() == 0
    s = model.toDebugString()
    assert "Tree 0" in s and "feature" in s
    assert model.treeWeights == [1.0] * 10
    assert model.totalNumNodes > 10


def test_rf_threshold_equality_consistent_with_training():
    """Rows whose feature value EQUALS a split threshold must take the same
    branch at predict time as at fit time (round-1 advisor finding: training
    partitions on x < edges[b] but inference used <=). Integer-valued,
    perfectly separable data makes every split land exactly on a data value."""
    rng = np.random.default_rng(0)
    X = rng.integers(0, 10, size=(800, 4)).astype(np.float32)
    y = (X[:, 0] >= 5).astype(np.float64)  # separable on an integer boundary
    df = DataFrame({"features": X, "label": y})
    m = RandomForestClassifier(
        featuresCol="features", labelCol="label", numTrees=1, maxDepth=4,
        bootstrap=False, seed=3, featureSubsetStrategy="all",
    ).fit(df)
    pred = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    acc = float((pred == y).mean())
    assert acc == 1.0, f"train accuracy {acc} < 1.0 on separable integer data"


def test_estimators_per_worker_split():
    """Tree division across ranks (reference tree.py:330-341: floor division
    with the remainder spread over the first ranks)."""
    from spark_rapids_ml_amd.models.tree import _estimators_per_worker

    assert _estimators_per_worker(10, 2) == [5, 5]
    assert _estimators_per_worker(10, 3) == [4, 3, 3]
    assert _estimators_per_worker(2, 4) == [1, 1, 0, 0]
    assert _estimators_per_worker(7, 1) == [7]
    assert sum(_estimators_per_worker(500, 8)) == 500


def test_resolve_max_features_strategies():
    """featureSubsetStrategy resolution (reference tree.py:93-135 mapping;
    Spark semantics: auto = sqrt for classification, onethird for
    regression)."""
    from spark_rapids_ml_amd.models.tree import _resolve_max_features

    d = 100
    assert _resolve_max_features("all", d, "classification") == d
    assert _resolve_max_features("sqrt", d, "classification") == 10
    assert _resolve_max_features("log2", d, "classification") == 6
    assert _resolve_max_features("onethird", d, "regression") == 33
    assert _resolve_max_features("auto", d, "classification") == 10
    assert _resolve_max_features("auto", d, "regression") == 33
    assert _resolve_max_features("0.5", d, "regression") == 50
    assert _resolve_max_features("25", d, "classification") == 25
    # results are clamped to [1, d]
    assert _resolve_max_features("0.0001", d, "regression") >= 1
