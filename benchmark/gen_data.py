"""Synthetic dataset generators (reference benchmark/gen_data.py +
gen_data_distributed.py): sklearn-style blobs / low-rank / regression /
classification / sparse-regression, generated DISTRIBUTED — every rank
produces its own row shard from a deterministic per-rank seed, optionally
writing sharded parquet.

CLI: python -m benchmark.gen_data blobs --num_rows 100000 --num_cols 300 \
        --output_dir /tmp/blobs [--dtype float32]
"""

from __future__ import annotations

import argparse
import sys
from typing import Optional, Tuple

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.parallel.context import get_comm


def _shard(num_rows: int, rank: int, world: int) -> int:
    return num_rows // world + (1 if rank < num_rows % world else 0)


def gen_blobs(
    num_rows: int, num_cols: int, centers: int = 20, cluster_std: float = 1.0,
    seed: int = 0, dtype: str = "float32",
) -> Tuple[np.ndarray, np.ndarray]:
    comm = get_comm()
    rng = np.random.default_rng(seed)
    C = rng.normal(scale=10.0, size=(centers, num_cols))
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    assign = rng_local.integers(0, centers, n_local)
    X = C[assign] + cluster_std * rng_local.normal(size=(n_local, num_cols))
    return X.astype(dtype), assign.astype(np.float64)


def gen_low_rank_matrix(
    num_rows: int, num_cols: int, effective_rank: int = 10, tail_strength: float = 0.5,
    seed: int = 0, dtype: str = "float32",
) -> np.ndarray:
    comm = get_comm()
    rng = np.random.default_rng(seed)
    # shared right factors; per-rank left factors
    sing = np.exp(-((np.arange(num_cols) / effective_rank) ** 2)) * (1 - tail_strength) + (
        tail_strength / (1 + np.arange(num_cols))
    )
    V = np.linalg.qr(rng.normal(size=(num_cols, num_cols)))[0]
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    U = rng_local.normal(size=(n_local, num_cols))
    X = (U * sing[None, :]) @ V.T
    return X.astype(dtype)


def gen_regression(
    num_rows: int, num_cols: int, n_informative: Optional[int] = None, noise: float = 1.0,
    bias: float = 0.5, effective_rank: Optional[int] = None, tail_strength: float = 0.5,
    seed: int = 0, dtype: str = "float32", return_coef: bool = False,
):
    """sklearn.make_regression semantics, rank-sharded (reference
    RegressionDataGen, gen_data_distributed.py:84-968): shared ground-truth
    coefficients from the GLOBAL seed, per-rank feature/noise draws; optional
    low-rank X via shared singular profile + right factors."""
    comm = get_comm()
    rng = np.random.default_rng(seed)
    ninf = n_informative or max(1, num_cols // 10)
    w = np.zeros(num_cols)
    w[rng.choice(num_cols, ninf, replace=False)] = rng.normal(scale=10.0, size=ninf)
    if effective_rank is not None:
        sing = np.exp(-((np.arange(num_cols) / effective_rank) ** 2)) * (
            1 - tail_strength
        ) + tail_strength / (1 + np.arange(num_cols))
        V = np.linalg.qr(rng.normal(size=(num_cols, num_cols)))[0]
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    X = rng_local.normal(size=(n_local, num_cols))
    if effective_rank is not None:
        X = (X * sing[None, :]) @ V.T
    y = X @ w + bias + noise * rng_local.normal(size=n_local)
    if return_coef:
        return X.astype(dtype), y.astype(np.float64), w
    return X.astype(dtype), y.astype(np.float64)


def gen_classification(
    num_rows: int, num_cols: int, n_classes: int = 2, n_informative: Optional[int] = None,
    n_redundant: Optional[int] = None, n_clusters_per_class: int = 1,
    class_sep: float = 1.0, flip_y: float = 0.01,
    seed: int = 0, dtype: str = "float32",
) -> Tuple[np.ndarray, np.ndarray]:
    """sklearn.make_classification semantics, rank-sharded (reference
    ClassificationDataGen): shared hypercube-ish centroids per (class,
    cluster) and a shared redundant-feature mixing matrix come from the
    GLOBAL seed; rows, cluster choices and label noise are per-rank."""
    comm = get_comm()
    rng = np.random.default_rng(seed)
    ninf = n_informative or max(2, num_cols // 10)
    nred = n_redundant if n_redundant is not None else 0
    assert ninf + nred <= num_cols, "n_informative + n_redundant > num_cols"
    # per-dim signal shrinks with sqrt(ninf) so the problem stays hard as
    # dimensionality grows (a solver must actually iterate)
    centroids = rng.normal(
        scale=2.0 * class_sep / np.sqrt(ninf),
        size=(n_classes, n_clusters_per_class, ninf),
    )
    mix = rng.normal(size=(ninf, nred)) if nred else None
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    y = rng_local.integers(0, n_classes, n_local)
    clust = rng_local.integers(0, n_clusters_per_class, n_local)
    X = rng_local.normal(size=(n_local, num_cols))
    X[:, :ninf] += centroids[y, clust]
    if mix is not None:
        # redundant features: linear combinations of the informative block
        X[:, ninf : ninf + nred] = X[:, :ninf] @ mix / np.sqrt(ninf)
    if flip_y > 0:
        flip = rng_local.random(n_local) < flip_y
        y[flip] = rng_local.integers(0, n_classes, int(flip.sum()))
    return X.astype(dtype), y.astype(np.float64)


def gen_sparse_regression(
    num_rows: int, num_cols: int, density: float = 0.1, n_informative: Optional[int] = None,
    noise: float = 1.0, seed: int = 0, dtype: str = "float32",
):
    import scipy.sparse as sp

    comm = get_comm()
    rng = np.random.default_rng(seed)
    ninf = n_informative or max(1, num_cols // 10)
    w = np.zeros(num_cols)
    w[rng.choice(num_cols, ninf, replace=False)] = rng.normal(scale=10.0, size=ninf)
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    X = sp.random(
        n_local, num_cols, density=density, format="csr", dtype=np.float64,
        random_state=np.random.RandomState(seed + comm.rank),
    )
    y = X @ w + noise * rng_local.normal(size=n_local)
    return X.astype(dtype), y.astype(np.float64)


def gen_sparse_classification_fast(
    num_rows: int, num_cols: int, nnz_per_row: int = 20, seed: int = 0, dtype: str = "float32",
):
    """Direct CSR construction (scipy.sparse.random is too slow past ~1e8
    nnz): fixed nnz per row, sorted column draws, synthetic linear labels."""
    import scipy.sparse as sp

    comm = get_comm()
    rng = np.random.default_rng(seed)
    w = rng.normal(size=num_cols)
    rng_local = np.random.default_rng(seed + 104729 * (comm.rank + 1))
    n_local = _shard(num_rows, comm.rank, comm.world_size)
    cols = np.sort(
        rng_local.integers(0, num_cols, size=(n_local, nnz_per_row), dtype=np.int32),
        axis=1,
    )
    data = rng_local.normal(size=(n_local, nnz_per_row)).astype(dtype)
    indptr = np.arange(0, (n_local + 1) * nnz_per_row, nnz_per_row, dtype=np.int64)
    X = sp.csr_matrix((data.ravel(), cols.ravel(), indptr), shape=(n_local, num_cols))
    margin = (data * w[cols]).sum(axis=1)
    y = (margin + 0.3 * rng_local.normal(size=n_local) > 0).astype(np.float64)
    return X, y


GENERATORS = {
    "blobs": gen_blobs,
    "low_rank_matrix": gen_low_rank_matrix,
    "regression": gen_regression,
    "classification": gen_classification,
    "sparse_regression": gen_sparse_regression,
    "sparse_classification_fast": gen_sparse_classification_fast,
}


def main() -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("kind", choices=sorted(GENERATORS))
    ap.add_argument("--num_rows", type=int, default=10000)
    ap.add_argument("--num_cols", type=int, default=100)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--dtype", default="float32")
    ap.add_argument("--output_dir", required=True)
    ap.add_argument("--centers", type=int, default=20)
    ap.add_argument("--n_classes", type=int, default=2)
    ap.add_argument("--density", type=float, default=0.1)
    args = ap.parse_args()

    kw = dict(num_rows=args.num_rows, num_cols=args.num_cols, seed=args.seed, dtype=args.dtype)
    if args.kind == "blobs":
        kw["centers"] = args.centers
    if args.kind == "classification":
        kw["n_classes"] = args.n_classes
    if args.kind == "sparse_regression":
        kw["density"] = args.density
    out = GENERATORS[args.kind](**kw)
    if isinstance(out, tuple):
        X, y = out
        df = DataFrame.from_numpy(X, y)
    else:
        df = DataFrame.from_numpy(out)
    df.write_parquet(args.output_dir)
    print(f"rank {get_comm().rank}: wrote {df.num_rows} rows to {args.output_dir}")


if __name__ == "__main__":
    main()
