"""Property-based invariants (hypothesis): the distributed evaluators rely on
sufficient statistics whose merge must be exactly order- and
partition-independent — any violation shows up as rank-count-dependent
metrics. These properties pin that down harder than example tests."""

import numpy as np
from hypothesis import given, settings, strategies as st

from spark_rapids_ml_amd.metrics import MulticlassMetrics, RegressionMetrics
from spark_rapids_ml_amd.metrics.RegressionMetrics import _SummarizerBuffer


@st.composite
def labels_preds(draw):
    n = draw(st.integers(min_value=2, max_value=200))
    seed = draw(st.integers(min_value=0, max_value=2**31 - 1))
    rng = np.random.default_rng(seed)
    lab = rng.integers(0, 4, size=n).astype(np.float64)
    pred = rng.integers(0, 4, size=n).astype(np.float64)
    split = draw(st.integers(min_value=0, max_value=n))
    return lab, pred, split


def _mc_from(lab, pred):
    conf = {}
    pairs = np.stack([lab, pred], axis=1)
    uniq, counts = np.unique(pairs, axis=0, return_counts=True)
    conf = {(float(a), float(b)): float(c) for (a, b), c in zip(uniq, counts)}
    return MulticlassMetrics.from_confusion(conf)


@settings(max_examples=40, deadline=None)
@given(labels_preds())
def test_multiclass_merge_is_partition_independent(data):
    lab, pred, split = data
    whole = _mc_from(lab, pred)
    parts = _mc_from(lab[:split], pred[:split]).merge(_mc_from(lab[split:], pred[split:]))
    for metric in ("f1", "accuracy", "weightedPrecision", "weightedRecall", "hammingLoss"):
        assert np.isclose(whole.evaluate(metric), parts.evaluate(metric), atol=1e-12)


@settings(max_examples=40, deadline=None)
@given(labels_preds())
def test_regression_merge_is_partition_independent(data):
    lab, pred, split = data
    lab = lab + 0.25  # avoid degenerate all-equal slices less often
    whole = RegressionMetrics.from_predictions(lab, pred)
    a = RegressionMetrics.from_predictions(lab[:split], pred[:split])
    b = RegressionMetrics.from_predictions(lab[split:], pred[split:])
    merged = a.merge(b)
    for metric in ("mse", "rmse", "mae", "r2", "var"):
        w = whole.evaluate(metric)
        m = merged.evaluate(metric)
        assert np.isclose(w, m, rtol=1e-9, atol=1e-9), (metric, w, m)


@settings(max_examples=30, deadline=None)
@given(
    st.integers(min_value=0, max_value=2**31 - 1),
    st.integers(min_value=1, max_value=50),
)
def test_summarizer_buffer_list_roundtrip(seed, n):
    rng = np.random.default_rng(seed)
    lab = rng.normal(size=n)
    pred = rng.normal(size=n)
    buf = RegressionMetrics.from_predictions(lab, pred)._buf
    restored = _SummarizerBuffer.from_list(buf.to_list())
    assert np.isclose(
        RegressionMetrics(restored).evaluate("mse"),
        RegressionMetrics(buf).evaluate("mse"),
    )


@settings(max_examples=30, deadline=None)
@given(st.integers(min_value=0, max_value=2**31 - 1))
def test_forest_persistence_roundtrip_predictions(seed):
    import tempfile

    from spark_rapids_ml_amd import RandomForestClassificationModel, RandomForestClassifier
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(seed)
    X = rng.normal(size=(120, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    model = RandomForestClassifier(numTrees=3, maxDepth=3, seed=seed % 1000).fit(
        DataFrame.from_numpy(X, y)
    )
    with tempfile.TemporaryDirectory() as tmp:
        model.write().overwrite().save(tmp + "/m")
        loaded = RandomForestClassificationModel.load(tmp + "/m")
    a = np.asarray(model.transform(DataFrame.from_numpy(X))["prediction"])
    b = np.asarray(loaded.transform(DataFrame.from_numpy(X))["prediction"])
    assert np.array_equal(a, b)
