"""Metrics parity vs sklearn (pattern: reference tests/test_metrics.py)."""

import numpy as np
import pytest
from sklearn import metrics as skm

from spark_rapids_ml_amd.metrics import MulticlassMetrics, RegressionMetrics
from spark_rapids_ml_amd.metrics.RegressionMetrics import _SummarizerBuffer


def _conf(y_true, y_pred):
    conf = {}
    for l, p in zip(y_true, y_pred):
        conf[(float(l), float(p))] = conf.get((float(l), float(p)), 0.0) + 1.0
    return conf


def test_multiclass_metrics_match_sklearn():
    rng = np.random.default_rng(0)
    y_true = rng.integers(0, 3, 500).astype(float)
    y_pred = np.where(rng.random(500) < 0.7, y_true, rng.integers(0, 3, 500)).astype(float)
    m = MulticlassMetrics.from_confusion(_conf(y_true, y_pred))
    assert np.isclose(m.evaluate("accuracy"), skm.accuracy_score(y_true, y_pred))
    assert np.isclose(m.evaluate("f1"), skm.f1_score(y_true, y_pred, average="weighted"))
    assert np.isclose(
        m.evaluate("weightedPrecision"),
        skm.precision_score(y_true, y_pred, average="weighted"),
    )
    assert np.isclose(
        m.evaluate("weightedRecall"), skm.recall_score(y_true, y_pred, average="weighted")
    )
    assert np.isclose(m.evaluate("hammingLoss"), skm.hamming_loss(y_true, y_pred))


def test_multiclass_merge_associative():
    rng = np.random.default_rng(1)
    y_true = rng.integers(0, 4, 300).astype(float)
    y_pred = rng.integers(0, 4, 300).astype(float)
    whole = MulticlassMetrics.from_confusion(_conf(y_true, y_pred))
    a = MulticlassMetrics.from_confusion(_conf(y_true[:100], y_pred[:100]))
    b = MulticlassMetrics.from_confusion(_conf(y_true[100:], y_pred[100:]))
    merged = a.merge(b)
    assert np.isclose(merged.evaluate("f1"), whole.evaluate("f1"))
    assert np.isclose(merged.evaluate("accuracy"), whole.evaluate("accuracy"))


def test_log_loss():
    rng = np.random.default_rng(2)
    y = rng.integers(0, 2, 200).astype(float)
    probs = rng.random((200, 2))
    probs = probs / probs.sum(axis=1, keepdims=True)
    ll_local = float(-np.log(np.clip(probs[np.arange(200), y.astype(int)], 1e-15, 1)).sum())
    m = MulticlassMetrics.from_confusion(_conf(y, probs.argmax(1).astype(float)), ll_local)
    assert np.isclose(m.evaluate("logLoss"), skm.log_loss(y, probs))


def test_regression_metrics_match_sklearn():
    rng = np.random.default_rng(3)
    y = rng.normal(size=400)
    pred = y + 0.3 * rng.normal(size=400)
    m = RegressionMetrics.from_predictions(y, pred)
    assert np.isclose(m.evaluate("mse"), skm.mean_squared_error(y, pred))
    assert np.isclose(m.evaluate("rmse"), np.sqrt(skm.mean_squared_error(y, pred)))
    assert np.isclose(m.evaluate("mae"), skm.mean_absolute_error(y, pred))
    assert np.isclose(m.evaluate("r2"), skm.r2_score(y, pred))


def test_regression_merge_matches_whole():
    rng = np.random.default_rng(4)
    y = rng.normal(size=300)
    pred = y + 0.2 * rng.normal(size=300)
    whole = RegressionMetrics.from_predictions(y, pred)
    a = RegressionMetrics.from_predictions(y[:120], pred[:120])
    b = RegressionMetrics.from_predictions(y[120:], pred[120:])
    merged = a.merge(b)
    for name in ("rmse", "mse", "mae", "r2"):
        assert np.isclose(merged.evaluate(name), whole.evaluate(name)), name


def test_summarizer_buffer_roundtrip():
    buf = _SummarizerBuffer([1.0, 2.0], [0.1, 0.2], [1.1, 2.2], [3.0, 4.0], 10)
    back = _SummarizerBuffer.from_list(buf.to_list())
    assert back.total_count == 10
    assert back._curr_mean == [1.0, 2.0]


def test_by_label_metrics():
    from sklearn.metrics import precision_score, recall_score

    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator

    rng = np.random.default_rng(0)
    lab = rng.integers(0, 3, size=500).astype(np.float64)
    pred = lab.copy()
    flip = rng.random(500) < 0.2
    pred[flip] = rng.integers(0, 3, size=flip.sum()).astype(np.float64)
    df = DataFrame({"label": lab, "prediction": pred})
    for cls in (0.0, 1.0, 2.0):
        p = MulticlassClassificationEvaluator(
            metricName="precisionByLabel", metricLabel=cls
        ).evaluate(df)
        r = MulticlassClassificationEvaluator(
            metricName="recallByLabel", metricLabel=cls
        ).evaluate(df)
        assert np.isclose(p, precision_score(lab, pred, labels=[cls], average=None)[0])
        assert np.isclose(r, recall_score(lab, pred, labels=[cls], average=None)[0])


def test_auc_matches_sklearn():
    from sklearn.metrics import roc_auc_score

    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import BinaryClassificationEvaluator

    rng = np.random.default_rng(0)
    y = rng.integers(0, 2, size=500).astype(np.float64)
    score = y + rng.normal(scale=1.2, size=500)  # informative but noisy
    raw = np.stack([-score, score], axis=1)
    df = DataFrame({"label": y, "rawPrediction": raw})
    auc = BinaryClassificationEvaluator().evaluate(df)
    assert np.isclose(auc, roc_auc_score(y, score), atol=1e-9)


def _spark_pr_auc(y: np.ndarray, score: np.ndarray) -> float:
    """Independent oracle for Spark BinaryClassificationMetrics.areaUnderPR:
    PR points at distinct thresholds descending, (0, p_first) start point,
    trapezoidal rule (reference semantics; Spark BinaryClassificationMetrics)."""
    pts = []
    for t in np.unique(score)[::-1]:
        sel = score >= t
        tp = float(((y == 1) & sel).sum())
        fp = float(((y != 1) & sel).sum())
        prec = tp / max(tp + fp, 1.0)
        rec = tp / max(float((y == 1).sum()), 1.0)
        pts.append((rec, prec))
    pts = [(0.0, pts[0][1])] + pts
    area = 0.0
    for (r0, p0), (r1, p1) in zip(pts[:-1], pts[1:]):
        area += (r1 - r0) * (p0 + p1) / 2.0
    return area


def test_area_under_pr_spark_semantics():
    from sklearn.metrics import average_precision_score

    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import BinaryClassificationEvaluator

    rng = np.random.default_rng(1)
    y = rng.integers(0, 2, size=400).astype(np.float64)
    score = y + rng.normal(scale=1.0, size=400)
    df = DataFrame({"label": y, "rawPrediction": np.stack([-score, score], axis=1)})
    pr = BinaryClassificationEvaluator(metricName="areaUnderPR").evaluate(df)
    assert np.isclose(pr, _spark_pr_auc(y, score), atol=1e-9)
    # trapezoid over a fine curve stays near (not equal to) sklearn AP
    assert abs(pr - average_precision_score(y, score)) < 0.02


def test_area_under_pr_tie_heavy():
    """Ties must collapse to one confusion point (Spark distinct-threshold
    semantics) before integration."""
    from spark_rapids_ml_amd.data import DataFrame
    from spark_rapids_ml_amd.evaluation import BinaryClassificationEvaluator

    rng = np.random.default_rng(7)
    y = rng.integers(0, 2, size=300).astype(np.float64)
    score = np.round(y + rng.normal(scale=1.0, size=300), 1)  # heavy ties
    df = DataFrame({"label": y, "rawPrediction": np.stack([-score, score], axis=1)})
    pr = BinaryClassificationEvaluator(metricName="areaUnderPR").evaluate(df)
    assert np.isclose(pr, _spark_pr_auc(y, score), atol=1e-9)
