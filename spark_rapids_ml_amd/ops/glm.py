"""GLM ops: fused score/loss/gradient passes for logistic & linear models.

Reference equivalent: LogisticRegressionMG's per-iteration forward+gradient
(SURVEY.md §2.3b classification.py:1046-1081). The HIP path keeps the GEMMs
(scores = X W^T, grad = R^T X) on MFMA and fuses the softmax/sigmoid loss +
residual into a single elementwise kernel between them.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import torch_ref
from .dispatch import hip_ops, use_hip


def logistic_grad_loss(
    X: torch.Tensor,
    y_idx: torch.Tensor,
    W: torch.Tensor,
    fit_intercept: bool,
    XT: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Local-shard gradient (sum, unscaled) and loss (sum) for binary
    (W: [1, d(+1)], sigmoid) or multinomial (W: [C, d(+1)], softmax).

    Dense X runs the GEMMs on MFMA (hipBLASLt) with the fused HIP
    softmax-residual kernel between them. Sparse CSR X (reference sparse
    path, classification.py:960-966) runs scores = X@Wᵀ and grad = (XᵀR)ᵀ
    through rocSPARSE SpMM; pass XT (the pre-transposed CSR) to avoid a
    transpose per iteration.
    """
    sparse = X.layout == torch.sparse_csr
    n = X.shape[0]
    d = X.shape[1]
    C = W.shape[0]
    coef = W[:, :d]
    hip_csr = sparse and use_hip(W) and C <= 32 and W.dtype == torch.float32
    if hip_csr:
        # own CSR kernels: thread-per-row forward, column-scatter gradient
        # (no pre-transposed CSR needed; rocSPARSE csrmm + the 400M-pair
        # transpose sort were the sparse-logreg wall)
        ext = hip_ops()
        scores = ext.csr_fwd(
            X.crow_indices(), X.col_indices(), X.values(),
            coef.T.contiguous(),
        )
    elif sparse:
        scores = torch.sparse.mm(X, coef.T.contiguous())
    else:
        scores = X @ coef.T
    if fit_intercept:
        scores = scores + W[:, d][None, :]

    if use_hip(scores) and y_idx.dtype == torch.int64:
        ext = hip_ops()
        resid, loss = ext.softmax_residual_loss(
            scores.contiguous().to(torch.float32), y_idx.contiguous()
        )
        resid = resid.to(scores.dtype)
    else:
        resid, loss = torch_ref.softmax_residual(scores, y_idx)

    if hip_csr:
        grad_coef = hip_ops().csr_grad(
            X.crow_indices(), X.col_indices(), X.values(),
            resid.to(torch.float32).contiguous(), d,
        )
    elif sparse:
        assert XT is not None, "sparse path requires pre-transposed CSR"
        grad_coef = torch.sparse.mm(XT, resid).T
    else:
        grad_coef = resid.T @ X
    if fit_intercept:
        grad = torch.cat([grad_coef, resid.sum(dim=0)[:, None]], dim=1)
    else:
        grad = grad_coef
    return grad, loss


def linear_grad_loss(
    X: torch.Tensor,
    y: torch.Tensor,
    w: torch.Tensor,
    fit_intercept: bool,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Squared-loss gradient (sum over local rows) for iterative linear
    solvers: resid = Xw(+b) - y; grad = [X^T resid, sum(resid)]; loss =
    0.5*||resid||^2."""
    n, d = X.shape
    coef = w[:d]
    pred = X @ coef
    if fit_intercept:
        pred = pred + w[d]
    resid = pred - y
    grad_coef = X.T @ resid
    if fit_intercept:
        grad = torch.cat([grad_coef, resid.sum().reshape(1)])
    else:
        grad = grad_coef
    loss = 0.5 * (resid * resid).sum()
    return grad, loss
