"""Pure-torch fp32/fp64 reference implementations of every HIP op.

These are the numerics oracles GPU tests compare the HIP kernels against,
and the execution path on CPU-only machines. They are written chunked so the
CPU test path stays memory-bounded on large inputs.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def kmeans_assign_reduce(
    X: torch.Tensor,
    C: torch.Tensor,
    x_sq: Optional[torch.Tensor] = None,
    chunk: int = 65536,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, float]:
    """Fused Lloyd step (assignment + per-center accumulation).

    Returns (labels int32 [n], sums f64 [k,d], counts f64 [k], inertia).
    Distance: squared euclidean via ||x||^2 + ||c||^2 - 2 x.c (the HIP kernel
    computes the same expansion with the -2XC^T term on MFMA).
    """
    n, d = X.shape
    k = C.shape[0]
    dev = X.device
    labels = torch.empty(n, dtype=torch.int32, device=dev)
    sums = torch.zeros((k, d), dtype=torch.float64, device=dev)
    counts = torch.zeros(k, dtype=torch.float64, device=dev)
    inertia = torch.zeros((), dtype=torch.float64, device=dev)
    if n == 0:
        return labels, sums, counts, 0.0
    c_sq = (C.to(torch.float32) ** 2).sum(dim=1)
    if x_sq is None:
        x_sq = (X.to(torch.float32) ** 2).sum(dim=1)
    for s in range(0, n, chunk):
        e = min(n, s + chunk)
        xb = X[s:e]
        dist = x_sq[s:e, None] + c_sq[None, :] - 2.0 * (xb @ C.T)
        md, lb = dist.min(dim=1)
        labels[s:e] = lb.to(torch.int32)
        sums.index_add_(0, lb, xb.to(torch.float64))
        counts.index_add_(0, lb, torch.ones_like(md, dtype=torch.float64))
        inertia += torch.clamp(md, min=0).to(torch.float64).sum()
    return labels, sums, counts, float(inertia.item())


def kmeans_predict(
    X: torch.Tensor, C: torch.Tensor, chunk: int = 65536
) -> torch.Tensor:
    n = X.shape[0]
    labels = torch.empty(n, dtype=torch.int32, device=X.device)
    if n == 0:
        return labels
    c_sq = (C**2).sum(dim=1)
    for s in range(0, n, chunk):
        e = min(n, s + chunk)
        xb = X[s:e]
        dist = c_sq[None, :] - 2.0 * (xb @ C.T)
        labels[s:e] = dist.argmin(dim=1).to(torch.int32)
    return labels


def gram(X: torch.Tensor) -> torch.Tensor:
    """X^T X in fp32 (or the input dtype if f64)."""
    return X.T @ X


def xty(X: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    return X.T @ y


def pairwise_sq_dists(
    X: torch.Tensor, Y: torch.Tensor
) -> torch.Tensor:
    x_sq = (X**2).sum(dim=1)
    y_sq = (Y**2).sum(dim=1)
    return torch.clamp(x_sq[:, None] + y_sq[None, :] - 2.0 * (X @ Y.T), min=0.0)


def knn_topk(
    Q: torch.Tensor,
    I: torch.Tensor,
    k: int,
    chunk: int = 8192,
    item_chunk_elems: int = 1 << 31,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Brute-force kNN of queries Q against items I: returns (dists [q,k],
    idx int64 [q,k]) with squared-euclid computed f32 then sqrt'ed.
    Items are chunked too (running top-k merge) so the distance block stays
    bounded — 10M items × an 8k query chunk would otherwise materialize
    hundreds of GB."""
    nq, ni = Q.shape[0], I.shape[0]
    k = min(k, ni)
    dists = torch.empty((nq, k), dtype=torch.float32, device=Q.device)
    idx = torch.empty((nq, k), dtype=torch.int64, device=Q.device)
    ichunk = max(k, min(ni, item_chunk_elems // max(1, chunk)))
    for s in range(0, nq, chunk):
        e = min(nq, s + chunk)
        qb = Q[s:e]
        q_sq = (qb**2).sum(dim=1)[:, None]
        best_d: torch.Tensor = None  # type: ignore[assignment]
        best_i: torch.Tensor = None  # type: ignore[assignment]
        for s2 in range(0, ni, ichunk):
            e2 = min(ni, s2 + ichunk)
            ib = I[s2:e2]
            d2 = q_sq + (ib**2).sum(dim=1)[None, :] - 2.0 * (qb @ ib.T)
            kk = min(k, e2 - s2)
            vals, ids = torch.topk(d2, kk, dim=1, largest=False)
            ids = ids + s2
            if best_d is None:
                best_d, best_i = vals, ids
            else:
                cat_d = torch.cat([best_d, vals], dim=1)
                cat_i = torch.cat([best_i, ids], dim=1)
                best_d, order = torch.topk(cat_d, min(k, cat_d.shape[1]), dim=1, largest=False)
                best_i = cat_i.gather(1, order)
        if best_d.shape[1] < k:  # ni < k edge
            pad = k - best_d.shape[1]
            best_d = torch.nn.functional.pad(best_d, (0, pad), value=float("inf"))
            best_i = torch.nn.functional.pad(best_i, (0, pad), value=-1)
        dists[s:e] = torch.sqrt(torch.clamp(best_d, min=0.0))
        idx[s:e] = best_i
    return dists, idx


def softmax_residual(
    scores: torch.Tensor, y_idx: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(resid [n,C], loss sum) — sigmoid when C==1 else softmax."""
    n, C = scores.shape[0], scores.shape[1] if scores.dim() > 1 else 1
    if C == 1:
        z = scores[:, 0]
        t = y_idx.to(z.dtype) * 2.0 - 1.0
        loss = torch.nn.functional.softplus(-t * z).sum()
        resid = (torch.sigmoid(z) - y_idx.to(z.dtype))[:, None]
    else:
        logp = torch.log_softmax(scores, dim=1)
        loss = -logp.gather(1, y_idx.view(-1, 1).to(torch.int64)).sum()
        resid = torch.exp(logp)
        resid.scatter_add_(
            1,
            y_idx.view(-1, 1).to(torch.int64),
            -torch.ones_like(y_idx, dtype=resid.dtype).view(-1, 1),
        )
    return resid, loss.reshape(())


def logistic_forward_grad(
    X: torch.Tensor,
    y_idx: torch.Tensor,
    W: torch.Tensor,
    fit_intercept: bool,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """One multinomial/binary logistic pass on the local shard.

    W is [C, d(+1)] with the intercept LAST column when fit_intercept.
    Binary (C==1): sigmoid/binomial loss in Spark's parametrization.
    Returns (grad [C, d(+1)] UNSCALED sum over local rows, loss scalar sum).
    """
    n, d = X.shape
    C = W.shape[0]
    coef = W[:, :d]
    scores = X @ coef.T
    if fit_intercept:
        scores = scores + W[:, d][None, :]
    if C == 1:
        z = scores[:, 0]
        # log(1+exp(-t*z)) with t in {-1,1}; y_idx in {0,1}
        t = y_idx.to(z.dtype) * 2.0 - 1.0
        loss = torch.nn.functional.softplus(-t * z).sum()
        p = torch.sigmoid(z)
        resid = (p - y_idx.to(z.dtype))[:, None]  # [n,1]
    else:
        logp = torch.log_softmax(scores, dim=1)
        loss = -logp.gather(1, y_idx.view(-1, 1).to(torch.int64)).sum()
        p = torch.exp(logp)
        p.scatter_add_(
            1,
            y_idx.view(-1, 1).to(torch.int64),
            -torch.ones_like(y_idx, dtype=p.dtype).view(-1, 1),
        )
        resid = p  # [n,C] = softmax - onehot
    grad_coef = resid.T @ X  # [C,d]
    if fit_intercept:
        grad = torch.cat([grad_coef, resid.sum(dim=0)[:, None]], dim=1)
    else:
        grad = grad_coef
    return grad, loss.reshape(())
