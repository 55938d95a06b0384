"""Pipeline + VectorAssembler bypass (pattern: reference
tests_no_import_change + pipeline tests)."""

import numpy as np
import pytest
from sklearn.datasets import make_classification

from spark_rapids_ml_amd import KMeans, LogisticRegression
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.pipeline import NoOpTransformer, Pipeline, VectorAssembler


def _scalar_df(n=300, d=4, seed=0):
    X, y = make_classification(n_samples=n, n_features=d, n_informative=3, n_redundant=0, random_state=seed)
    cols = {f"c{i}": X[:, i].astype(np.float64) for i in range(d)}
    cols["label"] = y.astype(np.float64)
    return DataFrame(cols), X, y


def test_vector_assembler():
    df, X, _ = _scalar_df()
    va = VectorAssembler(inputCols=["c0", "c1", "c2", "c3"], outputCol="features")
    out = va.transform(df)
    assert np.allclose(np.asarray(out["features"]), X.astype(np.float32), atol=1e-5)


def test_pipeline_bypasses_assembler():
    df, X, y = _scalar_df()
    va = VectorAssembler(inputCols=["c0", "c1", "c2", "c3"], outputCol="features")
    lr = LogisticRegression(maxIter=50)
    pipe = Pipeline(stages=[va, lr])
    model = pipe.fit(df)
    # assembler replaced by NoOp; estimator consumed featuresCols directly
    assert isinstance(model.stages[0], NoOpTransformer)
    assert lr.getFeaturesCols() == ["c0", "c1", "c2", "c3"]
    out = model.transform(df)
    acc = (np.asarray(out["prediction"]) == y).mean()
    assert acc > 0.65


def test_pipeline_without_assembler():
    df, X, y = _scalar_df()
    df = df.with_column("features", X.astype(np.float32))
    lr = LogisticRegression(maxIter=50)
    model = Pipeline(stages=[lr]).fit(df)
    out = model.transform(df)
    assert (np.asarray(out["prediction"]) == y).mean() > 0.65


def test_pipeline_persistence(tmp_model_path):
    import os

    from spark_rapids_ml_amd.pipeline import PipelineModel

    rng = np.random.default_rng(0)
    df = DataFrame({
        "a": rng.normal(size=200).astype(np.float32),
        "b": rng.normal(size=200).astype(np.float32),
    })
    pipe = Pipeline([VectorAssembler(["a", "b"]), KMeans(k=3, maxIter=10, seed=1)])
    pipe.save(os.path.join(tmp_model_path, "pipe"))
    loaded_pipe = Pipeline.load(os.path.join(tmp_model_path, "pipe"))
    assert loaded_pipe.getStages()[0].getInputCols() == ["a", "b"]
    assert loaded_pipe.getStages()[1].getOrDefault("k") == 3

    pm = pipe.fit(df)
    pm.save(os.path.join(tmp_model_path, "pm"))
    loaded = PipelineModel.load(os.path.join(tmp_model_path, "pm"))
    out = loaded.transform(df)
    assert np.array_equal(np.asarray(out["prediction"]), np.asarray(pm.transform(df)["prediction"]))


def _dist_pipeline(_):
    from spark_rapids_ml_amd import LogisticRegression
    from spark_rapids_ml_amd.parallel.context import get_comm
    from spark_rapids_ml_amd.pipeline import Pipeline, VectorAssembler

    comm = get_comm()
    rng = np.random.default_rng(0)
    a = rng.normal(size=1000).astype(np.float32)
    b = rng.normal(size=1000).astype(np.float32)
    lab = (a + 0.5 * b > 0).astype(np.float64)
    sl = slice(comm.rank, None, comm.world_size)
    df = DataFrame({"a": a[sl], "b": b[sl], "label": lab[sl]})
    pm = Pipeline([VectorAssembler(["a", "b"]), LogisticRegression(maxIter=40)]).fit(df)
    out = pm.transform(df)
    return np.asarray(pm.stages[-1].coefficients), (np.asarray(out["prediction"]) == lab[sl]).mean()


def test_pipeline_distributed():
    from tests.dist_utils import run_distributed

    results = run_distributed(_dist_pipeline, world_size=2, args=(None,))
    coef0, acc0 = results[0]
    coef1, acc1 = results[1]
    assert np.allclose(coef0, coef1)  # replicated model
    assert acc0 > 0.95 and acc1 > 0.95
