"""KMeans ops: fused assignment (MFMA distance + argmin) + accumulation.

HIP path (gfx950): `kmeans_assign` — LDS-tiled ‖x‖²+‖c‖²−2x·c with the
−2XCᵀ term on mfma_f32_32x32x2f32, fused per-row argmin (packed 64-bit
atomicMin) and block-reduced inertia; `label_accumulate` — vectorized
per-center sum/count scatter. Together these are the KMeansMG Lloyd-step
kernel family of the reference (clustering.py:381-415).
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
    """Returns (labels i32 [n], sums [k,d], counts [k], inertia)."""
    if use_hip(X) and X.dtype == torch.float32:
        ext = hip_ops()
        if x_sq is None:
            x_sq = _xsq(X)
        labels, _min_d, inertia = ext.kmeans_assign(
            X.contiguous(), C.contiguous(), x_sq.contiguous()
        )
        sums, counts = ext.label_accumulate(X.contiguous(), labels, C.shape[0])
        return labels, sums.to(torch.float64), counts.to(torch.float64), float(inertia.item())
    return torch_ref.kmeans_assign_reduce(X, C, x_sq)


def kmeans_predict(X: torch.Tensor, C: torch.Tensor) -> torch.Tensor:
    if use_hip(X) and X.dtype == torch.float32:
        ext = hip_ops()
        labels, _, _ = ext.kmeans_assign(X.contiguous(), C.contiguous(), _xsq(X))
        return labels
    return torch_ref.kmeans_predict(X, C)


def _xsq(X: torch.Tensor) -> torch.Tensor:
    return (X * X).sum(dim=1)
