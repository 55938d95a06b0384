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
    """Returns (labels i32 [n], sums [k,d], counts [k], inertia).

    Default GPU path: hipBLASLt GEMM for the C @ Xᵀ dot block (141 TF on the
    1M×3000×k=1000 headline shape) + the own `kmeans_argmin_kn` epilogue
    (one bandwidth-bound pass) + `label_accumulate`. The all-in-one MFMA
    `kmeans_assign` kernel (84 TF, profiles/README.md ladder) stays selected
    via SRML_KMEANS_VARIANT=fused."""
    import os

    if use_hip(X) and X.dtype == torch.float32:
        ext = hip_ops()
        if x_sq is None:
            x_sq = _xsq(X)
        n, k = X.shape[0], C.shape[0]
        fused = os.environ.get("SRML_KMEANS_VARIANT") == "fused"
        # Dispatch by k (measured on 100M x 128 / 1M x 3000, profiles/README):
        # - large k: [k, n] GEMM + kmeans_argmin_kn (47.6 vs 74.5 ms at
        #   k=1000 — the dot block is compute-shaped there)
        # - small k: X @ C^T tall-skinny GEMM + kmeans_argmin_nk wave-per-row
        #   epilogue (the fused kernel's 2-k-step pipeline never reaches
        #   steady state at small d, and the [k, n] skinny-m GEMM stalls:
        #   205 / 280 ms vs 66 ms at k=200)
        if not fused and 384 <= k and k * 4 <= 64 * 1024 and n > 0:
            labels, inertia = _assign_gemm(ext, X, C, x_sq, n, k)
        elif not fused and k < 384 and n > 0:
            labels, inertia = _assign_gemm_nk(ext, X, C, x_sq, n, k)
        else:
            labels, _min_d, inertia = ext.kmeans_assign(
                X.contiguous(), C.contiguous(), x_sq.contiguous()
            )
        sums, counts = ext.label_accumulate(X.contiguous(), labels, C.shape[0])
        return labels, sums.to(torch.float64), counts.to(torch.float64), float(inertia.item())
    return torch_ref.kmeans_assign_reduce(X, C, x_sq)


def _assign_gemm(
    ext, X: torch.Tensor, C: torch.Tensor, x_sq: torch.Tensor, n: int, k: int,
    max_dots_bytes: int = 8 << 30,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GEMM + argmin-epilogue assignment, row-chunked so the [k, chunk] dot
    block stays under `max_dots_bytes`."""
    c_sq = (C * C).sum(dim=1).contiguous()
    Ct = C.contiguous()
    rchunk = max(256, int(max_dots_bytes // max(1, k * 4)))
    if rchunk >= n:
        dots = torch.mm(Ct, X.T)
        labels, _min_d, inertia = ext.kmeans_argmin_kn(dots, x_sq, c_sq)
        return labels, inertia
    labels = torch.empty(n, dtype=torch.int32, device=X.device)
    inertia = torch.zeros(1, dtype=torch.float64, device=X.device)
    dots = torch.empty((k, min(rchunk, n)), dtype=torch.float32, device=X.device)
    for s in range(0, n, rchunk):
        e = min(n, s + rchunk)
        dv = dots if e - s == dots.shape[1] else torch.empty(
            (k, e - s), dtype=torch.float32, device=X.device
        )
        torch.mm(Ct, X[s:e].T, out=dv)
        lb, _md, it = ext.kmeans_argmin_kn(dv, x_sq[s:e].contiguous(), c_sq)
        labels[s:e] = lb
        inertia += it
    return labels, inertia


def _assign_gemm_nk(
    ext, X: torch.Tensor, C: torch.Tensor, x_sq: torch.Tensor, n: int, k: int,
    max_dots_bytes: int = 8 << 30,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Tall-skinny GEMM (X_chunk @ Cᵀ, row-major [chunk, k]) + wave-per-row
    argmin epilogue — the small-k assignment path."""
    c_sq = (C * C).sum(dim=1).contiguous()
    CT = C.t().contiguous()  # [d, k] so the GEMM B operand is plain N-layout
    rchunk = max(256, int(max_dots_bytes // max(1, k * 4)))
    if rchunk >= n:
        dots = torch.mm(X, CT)
        labels, _min_d, inertia = ext.kmeans_argmin_nk(dots, x_sq, c_sq)
        return labels, inertia
    labels = torch.empty(n, dtype=torch.int32, device=X.device)
    inertia = torch.zeros(1, dtype=torch.float64, device=X.device)
    dots = torch.empty((min(rchunk, n), k), dtype=torch.float32, device=X.device)
    for s in range(0, n, rchunk):
        e = min(n, s + rchunk)
        dv = dots if e - s == dots.shape[0] else torch.empty(
            (e - s, k), dtype=torch.float32, device=X.device
        )
        torch.mm(X[s:e], CT, out=dv)
        lb, _md, it = ext.kmeans_argmin_nk(dv, x_sq[s:e].contiguous(), c_sq)
        labels[s:e] = lb
        inertia += it
    return labels, inertia


def kmeans_predict(X: torch.Tensor, C: torch.Tensor) -> torch.Tensor:
    import os

    if use_hip(X) and X.dtype == torch.float32:
        ext = hip_ops()
        n, k = X.shape[0], C.shape[0]
        fused = os.environ.get("SRML_KMEANS_VARIANT") == "fused"
        if not fused and 384 <= k and k * 4 <= 64 * 1024 and n > 0:
            labels, _ = _assign_gemm(ext, X.contiguous(), C, _xsq(X), n, k)
            return labels
        if not fused and k < 384 and n > 0:
            labels, _ = _assign_gemm_nk(ext, X.contiguous(), C, _xsq(X), n, k)
            return labels
        labels, _, _ = ext.kmeans_assign(X.contiguous(), C.contiguous(), _xsq(X))
        return labels
    return torch_ref.kmeans_predict(X, C)


def _xsq(X: torch.Tensor) -> torch.Tensor:
    return (X * X).sum(dim=1)
