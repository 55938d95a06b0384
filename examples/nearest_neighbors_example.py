"""Exact and approximate kNN: brute-force over all ranks' items, IVF-Flat,
IVF-PQ (+exact refine) and the CAGRA-equivalent graph search.

Single process:  python examples/nearest_neighbors_example.py
Multi-GPU:       srml-amd-launch examples/nearest_neighbors_example.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from spark_rapids_ml_amd import ApproximateNearestNeighbors, NearestNeighbors
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.parallel.context import get_comm

comm = get_comm()
rng = np.random.default_rng(comm.rank)
# clustered items (ANN indexes partition space; uniform noise is the
# degenerate worst case for any IVF/graph index)
centers = rng.normal(size=(64, 64)).astype(np.float32) * 4
assign = rng.integers(0, 64, size=100_000)
items = centers[assign] + rng.normal(size=(100_000, 64)).astype(np.float32)
qassign = rng.integers(0, 64, size=1_000)
queries = centers[qassign] + rng.normal(size=(1_000, 64)).astype(np.float32)
item_df = DataFrame({"features": items})
query_df = DataFrame({"features": queries})

exact = NearestNeighbors(k=10).fit(item_df)
_, _, knn = exact.kneighbors(query_df)
if comm.rank == 0:
    print("exact:", np.asarray(knn["indices"]).shape)

params = {
    "ivfflat": {"nlist": 64, "nprobe": 8},
    "ivfpq": {"nlist": 64, "nprobe": 16, "refine_ratio": 4.0},
    "cagra": {"graph_degree": 32, "itopk_size": 128, "max_iterations": 24},
}
for algo in ("ivfflat", "ivfpq", "cagra"):
    ann = ApproximateNearestNeighbors(
        k=10, algorithm=algo, algoParams=params[algo]
    ).fit(item_df)
    _, _, aknn = ann.kneighbors(query_df)
    ei = np.asarray(knn["indices"])
    gi = np.asarray(aknn["indices"])
    recall = np.mean([len(set(e) & set(g)) / 10 for e, g in zip(ei, gi)])
    if comm.rank == 0:
        print(f"{algo}: mean recall@10 = {recall:.3f}")
