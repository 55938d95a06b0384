"""Executor-side utilities (reference utils.py equivalents)."""

from __future__ import annotations

import contextlib
import logging
import sys
from typing import Any, Optional, Tuple

import numpy as np
import torch

_LOGGERS = {}


def get_logger(cls: Any, level: int = logging.INFO) -> logging.Logger:
    """Per-class stderr logger (reference utils.py:555-576)."""
    name = cls if isinstance(cls, str) else getattr(cls, "__name__", str(cls))
    if name in _LOGGERS:
        return _LOGGERS[name]
    logger = logging.getLogger(f"spark_rapids_ml_amd.{name}")
    logger.setLevel(level)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(
            logging.Formatter("%(asctime)s %(levelname)s %(name)s: %(message)s")
        )
        logger.addHandler(h)
        logger.propagate = False
    _LOGGERS[name] = logger
    return logger


def standardize_dataset(
    X: torch.Tensor,
    comm,
    pdesc,
    with_mean: bool = True,
    with_std: bool = True,
    eps: float = 1e-30,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Distributed column standardization (reference utils.py:876-982):
    partial sums and sums-of-squares are all-reduced, every rank applies the
    same (mean, stddev). Returns (X_std, mean, stddev); stddev uses the
    n-1 (sample) denominator to match Spark's Summarizer.

    Unlike the reference (which allGathers JSON lists of partial sums,
    utils.py:923-935) the partials here ride one fused RCCL all-reduce of a
    2×(d+?) tensor.
    """
    n_total = pdesc.m
    d = X.shape[1]
    acc = torch.zeros((2, d), dtype=torch.float64, device=X.device)
    if X.shape[0] > 0:
        for s0 in range(0, X.shape[0], 1 << 18):  # f32 chunk sums, f64 acc
            xb = X[s0 : s0 + (1 << 18)]
            acc[0] += xb.sum(dim=0).to(torch.float64)
            acc[1] += (xb * xb).sum(dim=0).to(torch.float64)
    acc = comm.allreduce_t(acc)
    mean = acc[0] / n_total
    if n_total > 1:
        var = (acc[1] - n_total * mean * mean) / (n_total - 1)
    else:
        var = torch.zeros_like(mean)
    var = torch.clamp(var, min=0.0)
    std = torch.sqrt(var)
    Xs = X
    if with_mean:
        Xs = Xs - mean.to(X.dtype)
    if with_std:
        denom = torch.where(std > eps, std, torch.ones_like(std))
        Xs = Xs / denom.to(X.dtype)
    return Xs, mean.to(X.dtype), std.to(X.dtype)


def as_numpy(t: torch.Tensor) -> np.ndarray:
    return t.detach().cpu().numpy()


@contextlib.contextmanager
def annotate(name: str):
    """rocprof-visible named range (roctx via torch's nvtx shim on ROCm).
    The reference has no tracer integration; these ranges make
    fit/transform phases attributable in rocprofv3 --sys-trace timelines."""
    pushed = False
    try:
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_push(name)
            pushed = True
    except Exception:
        pushed = False
    try:
        yield
    finally:
        if pushed:
            try:
                torch.cuda.nvtx.range_pop()
            except Exception:
                pass
