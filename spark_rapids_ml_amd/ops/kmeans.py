"""KMeans ops: fused assignment+accumulation (the Lloyd-step hot kernel).

HIP kernel: LDS-tiled distance expansion ||x||^2+||c||^2-2x.c with the
-2XC^T term on MFMA (mfma_f32_16x16x4f32), fused argmin and per-center
sum/count accumulation — the reference's KMeansMG fit kernel family
(reference clustering.py:381-415 invocation).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import torch_ref
from .dispatch import hip_ops, use_hip


def kmeans_assign_reduce(
    X: torch.Tensor,
    C: torch.Tensor,
    x_sq: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, float]:
    """Returns (labels i32 [n], sums f64|f32 [k,d], counts [k], inertia)."""
    if use_hip(X):
        ext = hip_ops()
        labels, sums, counts, inertia = ext.kmeans_assign_reduce(
            X.contiguous(), C.contiguous(), x_sq if x_sq is not None else _xsq(X)
        )
        return labels, sums, counts, float(inertia)
    return torch_ref.kmeans_assign_reduce(X, C, x_sq)


def kmeans_predict(X: torch.Tensor, C: torch.Tensor) -> torch.Tensor:
    if use_hip(X):
        ext = hip_ops()
        return ext.kmeans_predict(X.contiguous(), C.contiguous())
    return torch_ref.kmeans_predict(X, C)


def _xsq(X: torch.Tensor) -> torch.Tensor:
    return (X * X).sum(dim=1)
