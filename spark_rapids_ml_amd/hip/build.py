"""Build the in-tree HIP extension for gfx950.

Usage: python -m spark_rapids_ml_amd.hip.build

Compiles hip/ops.hip with hipcc (PYTORCH_ROCM_ARCH=gfx950) through
torch.utils.cpp_extension and leaves `_hip_ops.so` next to this file so the
repo snapshot carries it to GPU boxes (no JIT cache dependency).
"""

from __future__ import annotations

import os
import shutil

HERE = os.path.dirname(os.path.abspath(__file__))


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "16")
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(HERE, "build")
    os.makedirs(build_dir, exist_ok=True)
    mod = load(
        name="_hip_ops",
        sources=[os.path.join(HERE, "ops.hip")],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=True,
    )
    built = os.path.join(build_dir, "_hip_ops.so")
    dest = os.path.join(HERE, "_hip_ops.so")
    if os.path.exists(built):
        shutil.copy2(built, dest)
    return dest


if __name__ == "__main__":
    path = build()
    print(f"built {path}")
