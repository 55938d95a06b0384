"""Regression metrics from mergeable moment buffers (reference
metrics/RegressionMetrics.py:30-267): each rank summarizes its residuals as
(count, mean, m2n, m2, l1-norm) per column; buffers merge associatively."""

from __future__ import annotations

import math
from typing import List


class _SummarizerBuffer:
    """Mergeable moments of (label, label-prediction) columns
    (reference RegressionMetrics.py:30-168: mean/m2n/m2/l1/total count)."""

    def __init__(
        self,
        mean: List[float],
        m2n: List[float],
        m2: List[float],
        l1: List[float],
        total_cnt: int,
    ):
        self._curr_mean = list(mean)
        self._curr_m2n = list(m2n)
        self._curr_m2 = list(m2)
        self._curr_l1 = list(l1)
        self._total_cnt = total_cnt
        self._num_cols = len(mean)

    @classmethod
    def from_arrays(cls, label, residual) -> "_SummarizerBuffer":
        import numpy as np

        cols = [np.asarray(label, dtype=np.float64), np.asarray(residual, dtype=np.float64)]
        n = len(cols[0])
        mean = [float(c.mean()) if n else 0.0 for c in cols]
        m2n = [float(((c - c.mean()) ** 2).sum()) if n else 0.0 for c in cols]
        m2 = [float((c**2).sum()) for c in cols]
        l1 = [float(np.abs(c).sum()) for c in cols]
        return cls(mean, m2n, m2, l1, n)

    def merge(self, other: "_SummarizerBuffer") -> "_SummarizerBuffer":
        if other._total_cnt == 0:
            return self
        if self._total_cnt == 0:
            return other
        n1, n2 = self._total_cnt, other._total_cnt
        tot = n1 + n2
        mean, m2n, m2, l1 = [], [], [], []
        for i in range(self._num_cols):
            d = other._curr_mean[i] - self._curr_mean[i]
            new_mean = self._curr_mean[i] + d * n2 / tot
            mean.append(new_mean)
            m2n.append(self._curr_m2n[i] + other._curr_m2n[i] + d * d * n1 * n2 / tot)
            m2.append(self._curr_m2[i] + other._curr_m2[i])
            l1.append(self._curr_l1[i] + other._curr_l1[i])
        return _SummarizerBuffer(mean, m2n, m2, l1, tot)

    @property
    def total_count(self) -> int:
        return self._total_cnt

    def to_list(self) -> list:
        return [self._curr_mean, self._curr_m2n, self._curr_m2, self._curr_l1, self._total_cnt]

    @classmethod
    def from_list(cls, lst: list) -> "_SummarizerBuffer":
        return cls(*lst)


class RegressionMetrics:
    """rmse/mse/r2/mae/explained variance from a merged buffer
    (reference RegressionMetrics.py:170-267). Column order: [label, residual]."""

    def __init__(self, buf: _SummarizerBuffer):
        self._buf = buf

    @classmethod
    def from_predictions(cls, label, prediction) -> "RegressionMetrics":
        import numpy as np

        lab = np.asarray(label, dtype=np.float64)
        pred = np.asarray(prediction, dtype=np.float64)
        return cls(_SummarizerBuffer.from_arrays(lab, lab - pred))

    def merge(self, other: "RegressionMetrics") -> "RegressionMetrics":
        return RegressionMetrics(self._buf.merge(other._buf))

    @property
    def _ss_res(self) -> float:
        return self._buf._curr_m2[1]

    @property
    def _ss_tot(self) -> float:
        return self._buf._curr_m2n[0]

    def mean_squared_error(self) -> float:
        return self._ss_res / max(1, self._buf.total_count)

    def root_mean_squared_error(self) -> float:
        return math.sqrt(self.mean_squared_error())

    def mean_absolute_error(self) -> float:
        return self._buf._curr_l1[1] / max(1, self._buf.total_count)

    def r2(self) -> float:
        return 1.0 - self._ss_res / self._ss_tot if self._ss_tot > 0 else 0.0

    def explained_variance(self) -> float:
        return self._buf._curr_m2n[1] / max(1, self._buf.total_count)

    def evaluate(self, metric_name: str) -> float:
        dispatch = {
            "rmse": self.root_mean_squared_error,
            "mse": self.mean_squared_error,
            "mae": self.mean_absolute_error,
            "r2": self.r2,
            "var": self.explained_variance,
        }
        if metric_name not in dispatch:
            raise ValueError(f"Unsupported metric {metric_name!r}")
        return dispatch[metric_name]()
