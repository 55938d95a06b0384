"""Evaluators (pyspark.ml.evaluation equivalents over the SPMD runtime).

The reference plugs Spark's evaluators into its single-pass
transform+evaluate (reference core.py:1572-1693, metrics/*); pyspark is not
a dependency here, so the evaluator classes themselves are provided with the
same names/params. Each evaluator computes per-rank sufficient statistics
and merges them with one allgather — no raw predictions cross ranks.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np

from .data import DataFrame
from .metrics import MulticlassMetrics, RegressionMetrics
from .params import HasLabelCol, HasPredictionCol, Param, Params, TypeConverters
from .parallel.context import get_comm


class Evaluator(Params):
    def evaluate(self, df: DataFrame) -> float:
        raise NotImplementedError

    def isLargerBetter(self) -> bool:
        return True


class RegressionEvaluator(Evaluator, HasLabelCol, HasPredictionCol):
    """rmse|mse|r2|mae|var (Spark RegressionEvaluator parity)."""

    metricName = Param("eval", "metricName", "rmse|mse|r2|mae|var.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._setDefault(metricName="rmse")
        self._set(**kwargs)

    def setMetricName(self, value: str) -> "RegressionEvaluator":
        return self._set(metricName=value)  # type: ignore[return-value]

    def setLabelCol(self, value: str) -> "RegressionEvaluator":
        return self._set(labelCol=value)  # type: ignore[return-value]

    def setPredictionCol(self, value: str) -> "RegressionEvaluator":
        return self._set(predictionCol=value)  # type: ignore[return-value]

    def isLargerBetter(self) -> bool:
        return self.getOrDefault("metricName") in ("r2", "var")

    def evaluate(self, df: DataFrame) -> float:
        comm = get_comm()
        lab = np.asarray(df[self.getOrDefault("labelCol")])
        pred = np.asarray(df[self.getOrDefault("predictionCol")])
        local = RegressionMetrics.from_predictions(lab, pred)
        bufs = comm.allgather_obj(local._buf.to_list())
        from .metrics.RegressionMetrics import _SummarizerBuffer

        merged = local
        first = True
        for b in bufs:
            rm = RegressionMetrics(_SummarizerBuffer.from_list(b))
            merged = rm if first else merged.merge(rm)
            first = False
        return merged.evaluate(self.getOrDefault("metricName"))


class MulticlassClassificationEvaluator(Evaluator, HasLabelCol, HasPredictionCol):
    """f1|accuracy|weighted*|hammingLoss|logLoss (Spark parity)."""

    metricName = Param("eval", "metricName", "metric.", TypeConverters.toString)
    probabilityCol = Param("eval", "probabilityCol", "for logLoss.", TypeConverters.toString)
    metricLabel = Param("eval", "metricLabel", "class for *ByLabel metrics.", TypeConverters.toFloat)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._setDefault(metricName="f1", probabilityCol="probability", metricLabel=0.0)
        self._set(**kwargs)

    def setMetricName(self, value: str) -> "MulticlassClassificationEvaluator":
        return self._set(metricName=value)  # type: ignore[return-value]

    def setLabelCol(self, value: str) -> "MulticlassClassificationEvaluator":
        return self._set(labelCol=value)  # type: ignore[return-value]

    def setPredictionCol(self, value: str) -> "MulticlassClassificationEvaluator":
        return self._set(predictionCol=value)  # type: ignore[return-value]

    def isLargerBetter(self) -> bool:
        return self.getOrDefault("metricName") not in ("hammingLoss", "logLoss")

    def evaluate(self, df: DataFrame) -> float:
        comm = get_comm()
        lab = np.asarray(df[self.getOrDefault("labelCol")], dtype=np.float64)
        pred = np.asarray(df[self.getOrDefault("predictionCol")], dtype=np.float64)
        conf: Dict = {}
        if len(lab):
            pairs = np.stack([lab, pred], axis=1)
            uniq, counts = np.unique(pairs, axis=0, return_counts=True)
            conf = {(float(a), float(b)): float(c) for (a, b), c in zip(uniq, counts)}
        log_loss = None
        if self.getOrDefault("metricName") == "logLoss":
            probs = np.asarray(df[self.getOrDefault("probabilityCol")], dtype=np.float64)
            eps = 1e-15
            pl = np.clip(probs[np.arange(len(lab)), lab.astype(int)], eps, 1.0)
            log_loss = float(-np.log(pl).sum())
        all_stats = comm.allgather_obj((list(conf.items()), log_loss))
        merged = MulticlassMetrics()
        for items, ll in all_stats:
            m = MulticlassMetrics.from_confusion({tuple(k): v for k, v in items}, ll)
            merged = merged.merge(m)
        return merged.evaluate(
            self.getOrDefault("metricName"), self.getOrDefault("metricLabel")
        )


class BinaryClassificationEvaluator(Evaluator, HasLabelCol):
    """areaUnderROC (global Mann-Whitney rank-sum) and areaUnderPR
    (precision-recall step interpolation, Spark semantics) over gathered
    scores."""

    metricName = Param("eval", "metricName", "areaUnderROC|areaUnderPR.", TypeConverters.toString)
    rawPredictionCol = Param("eval", "rawPredictionCol", "scores col.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._setDefault(metricName="areaUnderROC", rawPredictionCol="rawPrediction")
        self._set(**kwargs)

    def evaluate(self, df: DataFrame) -> float:
        comm = get_comm()
        lab = np.asarray(df[self.getOrDefault("labelCol")], dtype=np.float64)
        raw = np.asarray(df[self.getOrDefault("rawPredictionCol")])
        score = raw[:, 1] if raw.ndim == 2 else raw
        parts = comm.allgather_obj((score, lab))
        s = np.concatenate([p[0] for p in parts])
        l = np.concatenate([p[1] for p in parts])
        pos = s[l == 1.0]
        neg = s[l != 1.0]
        if len(pos) == 0 or len(neg) == 0:
            return 0.5
        if self.getOrDefault("metricName") == "areaUnderPR":
            # Spark BinaryClassificationMetrics semantics: PR curve at
            # DISTINCT score thresholds (ties merged into one confusion
            # point), a (0, p_first) start point, trapezoidal integration.
            order = np.argsort(-s, kind="stable")
            so = s[order]
            tp = np.cumsum(l[order] == 1.0)
            fp = np.cumsum(l[order] != 1.0)
            # last index of each distinct-score run = the curve points
            last = np.nonzero(np.diff(so, append=np.nan))[0]
            tp, fp = tp[last], fp[last]
            precision = tp / np.maximum(tp + fp, 1)
            recall = tp / len(pos)
            recall = np.concatenate([[0.0], recall])
            precision = np.concatenate([[precision[0]], precision])
            return float(np.trapz(precision, recall))
        from scipy.stats import rankdata

        ranks = rankdata(np.concatenate([pos, neg]))
        auc = (ranks[: len(pos)].sum() - len(pos) * (len(pos) + 1) / 2) / (len(pos) * len(neg))
        return float(auc)
