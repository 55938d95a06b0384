import sys, time, torch
n, d, k = 1_000_000, 3000, 1000
g = torch.Generator(device="cuda").manual_seed(0)
X = torch.randn(n, d, generator=g, device="cuda")
C = torch.randn(k, d, generator=g, device="cuda")
out_nk = torch.empty(n, k, device="cuda")
out_kn = torch.empty(k, n, device="cuda")
def t(fn, rep=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(rep): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/rep
t1 = t(lambda: torch.mm(X, C.T, out=out_nk))
t2 = t(lambda: torch.mm(C, X.T, out=out_kn))
fl = 2.0*n*d*k
print(f"mm [n,k]: {t1*1000:.1f} ms = {fl/t1/1e12:.1f} TF")
print(f"mm [k,n]: {t2*1000:.1f} ms = {fl/t2/1e12:.1f} TF")
from spark_rapids_ml_amd.ops import kmeans_assign_reduce
x_sq = (X*X).sum(1)
t3 = t(lambda: kmeans_assign_reduce(X, C, x_sq), rep=3)
print(f"current fused assign_reduce: {t3*1000:.1f} ms")
