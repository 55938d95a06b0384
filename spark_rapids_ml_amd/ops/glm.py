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
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Local-shard gradient (sum, unscaled) and loss (sum) for binary
    (W: [1, d(+1)], sigmoid) or multinomial (W: [C, d(+1)], softmax)."""
    if use_hip(X):
        ext = hip_ops()
        if hasattr(ext, "softmax_residual_loss"):
            n, d = X.shape
            C = W.shape[0]
            coef = W[:, :d]
            scores = X @ coef.T
            if fit_intercept:
                scores = scores + W[:, d][None, :]
            resid, loss = ext.softmax_residual_loss(scores.contiguous(), y_idx.contiguous())
            grad_coef = resid.T @ X
            if fit_intercept:
                grad = torch.cat([grad_coef, resid.sum(dim=0)[:, None]], dim=1)
            else:
                grad = grad_coef
            return grad, loss
    return torch_ref.logistic_forward_grad(X, y_idx, W, fit_intercept)


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
