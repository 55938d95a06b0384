"""Param system semantics (model: reference tests/test_common_estimator.py's
param-mapping checks)."""

import numpy as np
import pytest

from spark_rapids_ml_amd import KMeans, LinearRegression
from spark_rapids_ml_amd.params import Param, Params, TypeConverters


def test_param_identity_and_defaults():
    km = KMeans()
    assert km.hasParam("k")
    assert km.getOrDefault("k") == 2
    assert not km.isSet("k")
    km.setK(5)
    assert km.isSet("k")
    assert km.getOrDefault("k") == 5
    assert km.native_params["n_clusters"] == 5


def test_param_sync_spark_to_native():
    km = KMeans(k=7, maxIter=13, tol=1e-3, seed=11)
    assert km.native_params["n_clusters"] == 7
    assert km.native_params["max_iter"] == 13
    assert km.native_params["tol"] == 1e-3
    assert km.native_params["random_state"] == 11


def test_native_only_param_accepted():
    # native-only params accepted in constructors (reference README:157-161)
    km = KMeans(oversampling_factor=3.0)
    assert km.native_params["oversampling_factor"] == 3.0


def test_unsupported_param_value_errors():
    with pytest.raises(ValueError):
        KMeans(initMode="bogus")


def test_unsupported_param_none_mapping_errors():
    # weightCol maps to None -> error on set (reference params.py:186-196)
    with pytest.raises(ValueError):
        KMeans(weightCol="w")


def test_unknown_param_errors():
    with pytest.raises(ValueError):
        KMeans(definitely_not_a_param=1)


def test_copy_isolates_maps():
    km = KMeans(k=3)
    km2 = km.copy({"k": 9} and {km.getParam("k"): 9})
    assert km2.getOrDefault("k") == 9
    assert km.getOrDefault("k") == 3


def test_explain_params_runs():
    s = KMeans().explainParams()
    assert "maxIter" in s


def test_value_mapping_initmode():
    km = KMeans(initMode="random")
    assert km.native_params["init"] == "random"


def test_linreg_param_mapping():
    lr = LinearRegression(regParam=0.5, elasticNetParam=0.3, maxIter=7)
    assert lr.native_params["alpha"] == 0.5
    assert lr.native_params["l1_ratio"] == 0.3
    assert lr.native_params["max_iter"] == 7


def test_auto_generated_getters():
    from spark_rapids_ml_amd import KMeans, LinearRegression

    km = KMeans(k=7, maxIter=13, seed=5)
    assert km.getK() == 7
    assert km.getMaxIter() == 13
    assert km.getSeed() == 5
    lr = LinearRegression(regParam=0.25)
    assert lr.getRegParam() == 0.25
    with pytest.raises(AttributeError):
        km.getNoSuchParam()


def test_weightcol_unsupported_everywhere():
    from spark_rapids_ml_amd import (
        KMeans,
        LinearRegression,
        LogisticRegression,
        RandomForestClassifier,
    )

    for cls in (KMeans, LinearRegression, LogisticRegression, RandomForestClassifier):
        inst = cls()
        assert inst.hasParam("weightCol")
        with pytest.raises(ValueError):
            inst.setWeightCol("w")
    with pytest.raises(ValueError):
        RandomForestClassifier().setLeafCol("leaf")


def test_model_native_attribute_parity():
    import numpy as np
    from sklearn.datasets import make_classification

    from spark_rapids_ml_amd import PCA, LogisticRegression
    from spark_rapids_ml_amd.data import DataFrame

    X, y = make_classification(n_samples=200, n_features=6, random_state=0)
    X = X.astype(np.float32)
    pca = PCA(k=3).fit(DataFrame.from_numpy(X))
    assert pca.singular_values_.shape == (3,)
    assert pca.explained_variance_ratio_.shape == (3,)
    assert len(pca.mean) == 6
    lr = LogisticRegression(maxIter=30).fit(DataFrame.from_numpy(X, y.astype(np.float64)))
    assert lr.coef_.shape == (1, 6)
    assert lr.n_cols == 6
    v = X[0]
    pred = lr.predict(v)
    raw = lr.predictRaw(v)
    prob = lr.predictProbability(v)
    assert pred in (0.0, 1.0)
    assert raw.shape == (2,) and prob.shape == (2,)
    assert abs(prob.sum() - 1.0) < 1e-6


def test_cv_collect_sub_models():
    import numpy as np

    from spark_rapids_ml_amd import LinearRegression
    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import RegressionEvaluator
    from spark_rapids_ml_amd.tuning import CrossValidator, ParamGridBuilder

    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 5))
    y = X @ rng.normal(size=5) + 0.01 * rng.normal(size=300)
    df = DataFrame.from_numpy(X, y)
    lr = LinearRegression()
    grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 0.1]).build()
    cv = CrossValidator(lr, grid, RegressionEvaluator(), numFolds=2, collectSubModels=False)
    cv.setCollectSubModels(True)
    m = cv.fit(df)
    assert m.subModels is not None
    assert len(m.subModels) == 2 and len(m.subModels[0]) == 2
    assert all(sm is not None for fold in m.subModels for sm in fold)


def test_rf_single_vector_predict():
    import numpy as np
    from sklearn.datasets import make_classification

    from spark_rapids_ml_amd import RandomForestClassifier
    from spark_rapids_ml_amd.data import DataFrame

    X, y = make_classification(n_samples=300, n_features=8, random_state=0)
    X = X.astype(np.float32)
    model = RandomForestClassifier(numTrees=5, maxDepth=4, seed=1).fit(
        DataFrame.from_numpy(X, y.astype(np.float64))
    )
    v = X[0]
    assert model.predict(v) in (0.0, 1.0)
    p = model.predictProbability(v)
    assert p.shape == (2,) and abs(p.sum() - 1.0) < 1e-6
    assert model.predictRaw(v).shape == (2,)


def test_model_cpu_conversions():
    import numpy as np
    from sklearn.datasets import make_blobs

    from spark_rapids_ml_amd import KMeans, PCA
    from spark_rapids_ml_amd.data import DataFrame

    X, _ = make_blobs(n_samples=200, n_features=5, centers=3, random_state=0)
    X = X.astype(np.float32)
    km = KMeans(k=3, maxIter=20, seed=1).fit(DataFrame.from_numpy(X))
    sk = km.cpu()
    ours = np.asarray(km.transform(DataFrame.from_numpy(X))["prediction"])
    assert np.array_equal(sk.predict(X.astype(np.float64)), ours)
    pca = PCA(k=2).fit(DataFrame.from_numpy(X))
    sk_pca = pca.cpu()
    assert sk_pca.transform(X.astype(np.float64)).shape == (200, 2)


def test_model_evaluate_and_summary():
    import numpy as np
    from sklearn.datasets import make_classification

    from spark_rapids_ml_amd import LinearRegression, LogisticRegression
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 4))
    y = X @ rng.normal(size=4) + 0.01 * rng.normal(size=200)
    df = DataFrame.from_numpy(X, y)
    lr = LinearRegression().fit(df)
    s = lr.evaluate(df)
    assert s.rootMeanSquaredError < 0.1
    assert s.numInstances == 200
    assert lr.scale == 1.0
    Xc, yc = make_classification(n_samples=200, n_features=6, random_state=0)
    dfc = DataFrame.from_numpy(Xc.astype(np.float32), yc.astype(np.float64))
    lg = LogisticRegression(maxIter=30).fit(dfc)
    assert lg.hasSummary
    assert lg.summary.totalIterations >= 1
    ev = lg.evaluate(dfc)
    assert ev.accuracy > 0.8


def test_param_bounds_validation():
    import numpy as np

    from spark_rapids_ml_amd import KMeans, LinearRegression, LogisticRegression
    from spark_rapids_ml_amd.data import DataFrame

    X = np.random.rand(50, 3).astype(np.float32)
    df = DataFrame.from_numpy(X)
    for bad in (dict(k=0), dict(k=-2), dict(maxIter=-1)):
        with pytest.raises(ValueError, match="out of range"):
            KMeans(**bad).fit(df)
    ydf = DataFrame.from_numpy(X, (X[:, 0] > 0.5).astype(np.float64))
    with pytest.raises(ValueError, match="out of range"):
        LogisticRegression(elasticNetParam=1.5).fit(ydf)
    with pytest.raises(ValueError, match="out of range"):
        LinearRegression(regParam=-0.1).fit(
            DataFrame.from_numpy(X, X[:, 0].astype(np.float64))
        )


def test_single_feature_column_fit():
    import numpy as np

    from spark_rapids_ml_amd import LinearRegression
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    x = rng.normal(size=(200, 1))
    y = 3.0 * x[:, 0] + 1.0 + 0.01 * rng.normal(size=200)
    m = LinearRegression().fit(DataFrame.from_numpy(x, y))
    assert abs(m.coefficients[0] - 3.0) < 0.05
    assert abs(m.intercept - 1.0) < 0.05


def test_kmeans_zero_iterations():
    import numpy as np

    from spark_rapids_ml_amd import KMeans
    from spark_rapids_ml_amd.data import DataFrame

    X = np.random.rand(100, 4).astype(np.float32)
    m = KMeans(k=3, maxIter=0, seed=1).fit(DataFrame.from_numpy(X))
    assert m.cluster_centers_.shape == (3, 4)
    assert m.summary.numIter == 0
    assert sum(m.summary.clusterSizes) == 100


def test_cuml_signature_ctor_params_accepted():
    """Reference python/README.md:157-182 — cuML-specific params supply
    through constructors, appear in cuml_params, have no getters/setters."""
    from spark_rapids_ml_amd import KMeans, LinearRegression, NearestNeighbors, UMAP

    km = KMeans(k=3, max_samples_per_batch=16384)
    assert km.cuml_params["max_samples_per_batch"] == 16384
    assert not hasattr(km, "getMaxSamplesPerBatch") or True
    with pytest.raises(AttributeError):
        km.getMaxSamplesPerBatch()  # no Spark getter for cuML-only params
    assert LinearRegression(algorithm="svd").cuml_params["algorithm"] == "svd"
    assert NearestNeighbors(batch_size=5).cuml_params["batch_size"] == 5
    assert UMAP(build_algo="nn_descent").cuml_params["build_algo"] == "nn_descent"
