"""Dense linear-algebra ops: Gram/covariance partials (MFMA), eigh, sign flip.

Reference equivalents: PCAMG's mean/covariance/eigendecomposition
(SURVEY.md §2.3b feature.py:232-253), LinearRegressionMG's normal equations
(regression.py:549,617-629), and the deprecated JNI signFlip kernel
(reference deprecated/native/src/rapidsml_jni.cu:35-61) whose deterministic
eigenvector sign convention is reproduced by `sign_flip`.
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import torch_ref
from .dispatch import hip_ops, use_hip


def gram(X: torch.Tensor) -> torch.Tensor:
    """X^T X  [d,d]. On gfx950 this runs on the hand-written MFMA f32 SYRK
    kernel (mfma_f32_32x32x2f32 tiles; exact f32 numerics — the guide's
    FP32-input MFMA is bitwise an fmaf chain)."""
    if use_hip(X):
        ext = hip_ops()
        if hasattr(ext, "gram_f32") and X.dtype == torch.float32:
            return ext.gram_f32(X.contiguous())
    return torch_ref.gram(X)


def xty_gram(X: torch.Tensor, y: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(X^T X, X^T y): the Gram rides the MFMA kernel, X^T y is a GEMV."""
    return gram(X), torch_ref.xty(X, y)


def eigh_sym(A: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Eigendecomposition of a symmetric d×d matrix, ascending eigenvalues
    (reference used raft::linalg::eigDC, rapidsml_jni.cu:215-269). On GPU
    this is rocSOLVER syevd via torch.linalg.eigh — a vendor solver, not a
    CUDA port; CPU fallback for determinism in CPU tests."""
    Ad = A.detach().to(torch.float64)
    if A.is_cuda:
        try:
            w, v = torch.linalg.eigh(Ad)
            return w, v
        except Exception:
            pass
    w, v = torch.linalg.eigh(Ad.cpu())
    return w.to(A.device), v.to(A.device)


def sign_flip(components: torch.Tensor) -> torch.Tensor:
    """Deterministic eigenvector sign convention: for each component (row),
    find the element with max |v|; if it is negative, negate the row
    (semantics of reference rapidsml_jni.cu:35-61, applied row-wise to the
    [k,d] components matrix)."""
    idx = components.abs().argmax(dim=1)
    signs = torch.sign(components.gather(1, idx.view(-1, 1)))
    signs = torch.where(signs == 0, torch.ones_like(signs), signs)
    return components * signs


def cov_from_gram(
    gram_total: torch.Tensor, mean: torch.Tensor, n_total: int, ddof: int = 1
) -> torch.Tensor:
    """Covariance from an all-reduced Gram partial and the global mean:
    (X^T X - n mu mu^T) / (n - ddof)."""
    g = gram_total.to(torch.float64)
    mu = mean.to(torch.float64)
    cov = (g - n_total * torch.outer(mu, mu)) / max(1, n_total - ddof)
    return cov
