"""Driver-side metric computation from mergeable per-rank sufficient
statistics (reference metrics/, 565 LoC)."""

from .MulticlassMetrics import MulticlassMetrics
from .RegressionMetrics import RegressionMetrics, _SummarizerBuffer


class EvalMetricInfo:
    """Evaluation request descriptor (reference metrics/__init__.py:30-40)."""

    def __init__(self, eval_metric_name: str = "", **kwargs):
        self.eval_metric_name = eval_metric_name
        for k, v in kwargs.items():
            setattr(self, k, v)


__all__ = ["MulticlassMetrics", "RegressionMetrics", "_SummarizerBuffer", "EvalMetricInfo"]
