"""k-NN ops: fused MFMA distance + in-LDS top-k selection.

HIP path (gfx950): `knn_select` streams 128-item tiles against 64-query
blocks, fusing the ||q-i||^2 expansion (MFMA) with per-query top-k candidate
pools held in LDS — the [nq, ni] distance matrix is never materialized
(reference NearestNeighborsMG tiled distance + top-k).
Falls back to the chunked torch path for k > 64 or CPU.
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import torch_ref
from .dispatch import hip_ops, use_hip


def knn_topk(Q: torch.Tensor, I: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """(dists [q,k] euclidean, idx int64 [q,k]) of queries against items.

    Default GPU path: hipBLASLt GEMM for the dot block + the OWN
    `knn_merge_topk` kernel, which fuses the ||q-i||^2 expansion with a
    running per-query top-k — the [q, chunk] distance matrix is never
    materialized and each GEMM output byte is read exactly once (the old
    torch path wrote d2, re-read it with multi-pass torch.topk and merged
    chunks with cat+topk+gather). SRML_KNN_KERNEL=1 keeps the all-in-one
    MFMA knn_select kernel (measured slower at 1M items; profiles/README.md),
    SRML_KNN_KERNEL=0 forces the torch reference path."""
    import os

    mode = os.environ.get("SRML_KNN_KERNEL", "")
    k_eff = min(k, I.shape[0])
    if mode == "1" and use_hip(Q, I) and k_eff <= 64 and Q.dtype == torch.float32:
        ext = hip_ops()
        d2, idx = ext.knn_select(Q.contiguous(), I.contiguous(), k_eff)
        return torch.sqrt(torch.clamp(d2, min=0.0)), idx
    if (
        mode != "0"
        and use_hip(Q, I)
        and k_eff == k
        and k <= 64
        and Q.dtype == torch.float32
        and I.dtype == torch.float32
        and Q.shape[0] > 0
    ):
        return _knn_topk_gemm_select(Q.contiguous(), I.contiguous(), k)
    return torch_ref.knn_topk(Q, I, k)


def _knn_topk_gemm_select(
    Q: torch.Tensor, I: torch.Tensor, k: int, chunk_elems: int = 1 << 29
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Library GEMM + own running-top-k selection kernel (see knn_topk)."""
    ext = hip_ops()
    nq, ni = Q.shape[0], I.shape[0]
    dev = Q.device
    q_sq = (Q * Q).sum(dim=1).contiguous()
    i_sq = (I * I).sum(dim=1).contiguous()
    # 64 candidate slots per query; lanes >= k carry a -inf sentinel so they
    # never own tau (the running worst kept distance) inside the kernel
    best_d = torch.full((nq, 64), float("inf"), dtype=torch.float32, device=dev)
    if k < 64:
        best_d[:, k:] = float("-inf")
    best_i = torch.full((nq, 64), -1, dtype=torch.int64, device=dev)

    # multiple of 64 so every non-final chunk keeps the kernel's float4 path
    # (row base 16B-aligned requires chunk width % 4 == 0)
    ichunk = max(256, (min(ni, chunk_elems // max(1, nq)) // 64) * 64)
    G = torch.empty((nq, min(ichunk, ni)), dtype=torch.float32, device=dev)
    for s in range(0, ni, ichunk):
        e = min(ni, s + ichunk)
        # the kernel wants a contiguous [nq, e-s] block (row stride == e-s)
        Gv = G if e - s == G.shape[1] else torch.empty(
            (nq, e - s), dtype=torch.float32, device=dev
        )
        torch.mm(Q, I[s:e].T, out=Gv)
        ext.knn_merge_topk(Gv, q_sq, i_sq[s:e].contiguous(), s, best_d, best_i)

    if k < 64:
        best_d[:, k:] = float("inf")  # dead slots sort last
    d_sorted, order = torch.sort(best_d, dim=1)
    i_sorted = best_i.gather(1, order)
    return (
        torch.sqrt(torch.clamp(d_sorted[:, :k], min=0.0)),
        i_sorted[:, :k].contiguous(),
    )
