"""UMAP 1M fit + kNN gemm-select for rocprof attribution."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch

which = sys.argv[1] if len(sys.argv) > 1 else "umap"
if which == "umap":
    from spark_rapids_ml_amd import UMAP
    from spark_rapids_ml_amd.data import DataFrame
    X = np.random.default_rng(0).normal(size=(1_000_000, 256)).astype(np.float32)
    t0 = time.perf_counter()
    UMAP(n_epochs=200).fit(DataFrame.from_numpy(X))
    torch.cuda.synchronize()
    print(f"umap fit {time.perf_counter()-t0:.2f}s")
else:
    from spark_rapids_ml_amd.ops.knn import _knn_topk_gemm_select
    g = torch.Generator(device="cuda").manual_seed(0)
    Q = torch.randn(10000, 768, generator=g, device="cuda")
    I = torch.randn(1_000_000, 768, generator=g, device="cuda")
    _knn_topk_gemm_select(Q, I, 64)  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        _knn_topk_gemm_select(Q, I, 64)
    torch.cuda.synchronize()
    print(f"knn gemm+select x3 {time.perf_counter()-t0:.3f}s")
