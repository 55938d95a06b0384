"""HIP extension loading & dispatch policy.

The extension is built IN-TREE (spark_rapids_ml_amd/hip/build.py ->
spark_rapids_ml_amd/hip/_hip_ops.so) so the .so travels with the repo
snapshot to GPU boxes. Loading is lazy; on a CUDA/ROCm-visible process the
HIP path is mandatory unless SRML_ALLOW_TORCH_FALLBACK=1.
"""

from __future__ import annotations

import importlib
import os
from typing import Any, Optional

import torch

_EXT: Optional[Any] = None
_TRIED = False


def _load() -> Optional[Any]:
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        from ..hip import _hip_ops  # type: ignore

        _EXT = _hip_ops
    except ImportError:
        # direct .so load (extension built as a plain shared object)
        here = os.path.join(os.path.dirname(__file__), "..", "hip")
        so = None
        if os.path.isdir(here):
            for f in os.listdir(here):
                if f.startswith("_hip_ops") and f.endswith(".so"):
                    so = os.path.join(here, f)
                    break
        if so is not None:
            spec = importlib.util.spec_from_file_location("_hip_ops", so)
            if spec and spec.loader:
                mod = importlib.util.module_from_spec(spec)
                try:
                    spec.loader.exec_module(mod)
                    _EXT = mod
                except Exception:
                    _EXT = None
    return _EXT


def has_hip_ops() -> bool:
    return _load() is not None


def hip_ops() -> Any:
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "HIP extension _hip_ops not built. Run "
            "`python -m spark_rapids_ml_amd.hip.build` (hipcc, gfx950)."
        )
    return ext


def require_hip_ops() -> Any:
    """On a GPU machine the HIP kernels must be the code that runs."""
    if os.environ.get("SRML_ALLOW_TORCH_FALLBACK") == "1":
        return _load()
    return hip_ops()


def use_hip(*tensors: torch.Tensor) -> bool:
    """True when tensors are on a ROCm device and the extension must be used."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if os.environ.get("SRML_ALLOW_TORCH_FALLBACK") == "1" and not has_hip_ops():
        return False
    return True
