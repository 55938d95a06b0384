"""Numeric ops: hand-written HIP/CDNA4 kernels with torch (CPU) references.

Layer L2 of SURVEY.md — the cuML/cuVS replacement. Each op has:
- a HIP implementation compiled for gfx950 (in-tree extension `_hip_ops`),
  used whenever tensors live on a ROCm device;
- a pure-torch reference used on CPU and as the numerics oracle in tests.

On a GPU box the HIP extension is REQUIRED for ops that have kernels: a
missing extension raises instead of silently falling back to eager torch
(set SRML_ALLOW_TORCH_FALLBACK=1 to override for debugging only).
"""

from .dispatch import has_hip_ops, hip_ops, require_hip_ops, use_hip
from . import torch_ref
from .kmeans import kmeans_assign_reduce, kmeans_predict
from .linalg import gram, xty_gram, eigh_sym, sign_flip
from .glm import logistic_grad_loss, linear_grad_loss
from .knn import knn_topk

__all__ = [
    "has_hip_ops",
    "hip_ops",
    "require_hip_ops",
    "use_hip",
    "torch_ref",
    "kmeans_assign_reduce",
    "kmeans_predict",
    "gram",
    "xty_gram",
    "eigh_sym",
    "sign_flip",
    "logistic_grad_loss",
    "knn_topk",
    "linear_grad_loss",
]
