import sys, time
sys.path.insert(0, "/root/repo")
import torch
from spark_rapids_ml_amd.ops.dispatch import hip_ops
ext = hip_ops()
g = torch.Generator(device="cuda").manual_seed(0)
def t(n, d, k, tag):
    X = torch.randn(n, d, generator=g, device="cuda")
    lab = torch.randint(0, k, (n,), generator=g, device="cuda", dtype=torch.int32)
    ext.label_accumulate(X, lab, k); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3): ext.label_accumulate(X, lab, k)
    torch.cuda.synchronize()
    ms = (time.perf_counter()-t0)/3*1000
    gb = n*d*4/1e9
    print(f"{tag}: n={n} d={d} k={k} -> {ms:.1f} ms  ({gb/ms*1000:.2f} GB/s ... {gb:.0f} GB)", flush=True)
    del X, lab; torch.cuda.empty_cache()
t(100_000_000, 128, 200, "baseline")
t(100_000_000, 128, 1,   "k1_contention")
t(200_000_000, 64, 200,  "2x_rows_same_bytes")
t(50_000_000, 128, 200,  "half")
t(100_000_000, 128, 16,  "k16_small_lds")
