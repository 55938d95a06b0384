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

    A/B on MI355X (10k/100k queries x 1M items x 768, k=64): the fused
    kernel runs 0.86/6.1 s vs 0.25/2.5 s for hipBLASLt GEMM + chunked
    torch.topk — the library-GEMM path wins until the fused kernel gets
    the full pipelining treatment, so it is the default; set
    SRML_KNN_KERNEL=1 to route through knn_select (profiles/README.md)."""
    import os

    k_eff = min(k, I.shape[0])
    if (
        os.environ.get("SRML_KNN_KERNEL") == "1"
        and use_hip(Q, I)
        and k_eff <= 64
        and Q.dtype == torch.float32
    ):
        ext = hip_ops()
        d2, idx = ext.knn_select(Q.contiguous(), I.contiguous(), k_eff)
        return torch.sqrt(torch.clamp(d2, min=0.0)), idx
    return torch_ref.knn_topk(Q, I, k)
