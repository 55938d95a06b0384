"""Multiclass metrics from confusion counts (reference
metrics/MulticlassMetrics.py:34-180): every metric is computed from the
mergeable (label, prediction) -> count map plus an optional summed log-loss,
so ranks only exchange tiny dictionaries."""

from __future__ import annotations

from typing import Dict, Optional, Tuple


class MulticlassMetrics:
    SUPPORTED_MULTI_CLASS_METRIC_NAMES = [
        "f1",
        "accuracy",
        "weightedPrecision",
        "weightedRecall",
        "weightedTruePositiveRate",
        "weightedFalsePositiveRate",
        "weightedFMeasure",
        "truePositiveRateByLabel",
        "falsePositiveRateByLabel",
        "precisionByLabel",
        "recallByLabel",
        "fMeasureByLabel",
        "hammingLoss",
        "logLoss",
    ]

    def __init__(
        self,
        tp: Optional[Dict[float, float]] = None,
        fp: Optional[Dict[float, float]] = None,
        label: Optional[Dict[float, float]] = None,
        label_count: int = 0,
        log_loss: Optional[float] = None,
    ):
        self._tp_by_class = tp or {}
        self._fp_by_class = fp or {}
        self._label_count_by_class = label or {}
        self._label_count = label_count
        self._log_loss = log_loss

    @classmethod
    def from_confusion(
        cls, conf: Dict[Tuple[float, float], float], log_loss: Optional[float] = None
    ) -> "MulticlassMetrics":
        """conf: {(label, prediction): count}."""
        tp: Dict[float, float] = {}
        fp: Dict[float, float] = {}
        label: Dict[float, float] = {}
        total = 0.0
        for (lab, pred), c in conf.items():
            total += c
            label[lab] = label.get(lab, 0.0) + c
            tp.setdefault(lab, 0.0)
            fp.setdefault(pred, 0.0)
            if lab == pred:
                tp[lab] = tp.get(lab, 0.0) + c
            else:
                fp[pred] = fp.get(pred, 0.0) + c
        return cls(tp, fp, label, int(total), log_loss)

    def merge(self, other: "MulticlassMetrics") -> "MulticlassMetrics":
        def madd(a, b):
            out = dict(a)
            for k, v in b.items():
                out[k] = out.get(k, 0.0) + v
            return out

        ll = None
        if self._log_loss is not None or other._log_loss is not None:
            ll = (self._log_loss or 0.0) + (other._log_loss or 0.0)
        return MulticlassMetrics(
            madd(self._tp_by_class, other._tp_by_class),
            madd(self._fp_by_class, other._fp_by_class),
            madd(self._label_count_by_class, other._label_count_by_class),
            self._label_count + other._label_count,
            ll,
        )

    def _precision(self, label: float) -> float:
        tp = self._tp_by_class.get(label, 0.0)
        fp = self._fp_by_class.get(label, 0.0)
        return 0.0 if (tp + fp) == 0 else tp / (tp + fp)

    def _recall(self, label: float) -> float:
        cnt = self._label_count_by_class.get(label, 0.0)
        return 0.0 if cnt == 0 else self._tp_by_class.get(label, 0.0) / cnt

    def _f_measure(self, label: float, beta: float = 1.0) -> float:
        p = self._precision(label)
        r = self._recall(label)
        b2 = beta * beta
        return 0.0 if (p + r) == 0 else (1 + b2) * p * r / (b2 * p + r)

    def false_positive_rate(self, label: float) -> float:
        fp = self._fp_by_class.get(label, 0.0)
        neg = self._label_count - self._label_count_by_class.get(label, 0.0)
        return 0.0 if neg == 0 else fp / neg

    def weighted_fmeasure(self, beta: float = 1.0) -> float:
        return sum(
            self._f_measure(k, beta) * v / self._label_count
            for k, v in self._label_count_by_class.items()
        )

    def accuracy(self) -> float:
        return sum(self._tp_by_class.values()) / self._label_count

    def weighted_precision(self) -> float:
        return sum(
            self._precision(k) * v / self._label_count
            for k, v in self._label_count_by_class.items()
        )

    def weighted_recall(self) -> float:
        return sum(
            self._recall(k) * v / self._label_count
            for k, v in self._label_count_by_class.items()
        )

    def weighted_true_positive_rate(self) -> float:
        return self.weighted_recall()

    def weighted_false_positive_rate(self) -> float:
        return sum(
            self.false_positive_rate(k) * v / self._label_count
            for k, v in self._label_count_by_class.items()
        )

    def hamming_loss(self) -> float:
        return 1.0 - self.accuracy()

    def log_loss(self) -> float:
        assert self._log_loss is not None, "log loss was not accumulated"
        return self._log_loss / self._label_count

    def evaluate(self, metric_name: str, metric_label: float = 0.0) -> float:
        dispatch = {
            "truePositiveRateByLabel": lambda: self._recall(metric_label),
            "falsePositiveRateByLabel": lambda: self.false_positive_rate(metric_label),
            "precisionByLabel": lambda: self._precision(metric_label),
            "recallByLabel": lambda: self._recall(metric_label),
            "fMeasureByLabel": lambda: self._f_measure(metric_label),
            "f1": lambda: self.weighted_fmeasure(),
            "accuracy": self.accuracy,
            "weightedPrecision": self.weighted_precision,
            "weightedRecall": self.weighted_recall,
            "weightedTruePositiveRate": self.weighted_true_positive_rate,
            "weightedFalsePositiveRate": self.weighted_false_positive_rate,
            "weightedFMeasure": lambda: self.weighted_fmeasure(),
            "hammingLoss": self.hamming_loss,
            "logLoss": self.log_loss,
        }
        if metric_name not in dispatch:
            raise ValueError(f"Unsupported metric {metric_name!r}")
        return dispatch[metric_name]()
