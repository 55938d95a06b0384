"""Standalone timing harness for the KMeans assign kernel variants.

Usage (on a GPU box):  [SRML_KMEANS_VARIANT=b|2|3|s] python tools/time_assign.py
Measures the 1M x 3000, k=1000 assign (the bench.py hot kernel) and checks
labels against a torch cdist argmin on a slice.
"""
import sys

import torch

sys.path.insert(0, ".")
from spark_rapids_ml_amd.ops.dispatch import hip_ops

ext = hip_ops()
dev = torch.device("cuda:0")
n, d, k = 1_000_000, 3000, 1000
g = torch.Generator(device=dev).manual_seed(0)
X = torch.randn(n, d, device=dev, generator=g)
C = X[:k].clone()
xsq = (X * X).sum(dim=1)
for _ in range(3):
    ext.kmeans_assign(X, C, xsq)
torch.cuda.synchronize()
s = torch.cuda.Event(enable_timing=True)
e = torch.cuda.Event(enable_timing=True)
s.record()
iters = 10
for _ in range(iters):
    labels, md, inertia = ext.kmeans_assign(X, C, xsq)
e.record()
torch.cuda.synchronize()
ms = s.elapsed_time(e) / iters
sl = slice(0, 2048)
dref = torch.cdist(X[sl].float(), C.float()) ** 2
lref = dref.argmin(dim=1)
match = (labels[sl].long() == lref).float().mean().item()
print(f"ASSIGN_MS {ms:.2f}  label_match {match:.4f}")
