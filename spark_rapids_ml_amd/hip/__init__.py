"""HIP/CDNA4 native kernels (gfx950). Built in-tree by `python -m
spark_rapids_ml_amd.hip.build` -> `_hip_ops*.so` next to this file."""
