"""Minimal label_accumulate-only workload for PMC collection (km100m shape)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_rapids_ml_amd.ops.dispatch import hip_ops

ext = hip_ops()
g = torch.Generator(device="cuda").manual_seed(0)
n, d, k = 20_000_000, 128, 200
X = torch.randn(n, d, generator=g, device="cuda")
labels = torch.randint(0, k, (n,), generator=g, device="cuda", dtype=torch.int32)
ext.label_accumulate(X, labels, k)
torch.cuda.synchronize()
for _ in range(3):
    ext.label_accumulate(X, labels, k)
torch.cuda.synchronize()
print("done")
