"""DBSCAN vs sklearn (pattern: reference tests/test_dbscan.py)."""

import numpy as np
import pytest
from sklearn.cluster import DBSCAN as SkDBSCAN
from sklearn.datasets import make_blobs
from sklearn.metrics import adjusted_rand_score

from spark_rapids_ml_amd import DBSCAN
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _data(n=300, seed=0):
    X, y = make_blobs(n_samples=n, n_features=4, centers=3, cluster_std=0.4, random_state=seed)
    return X.astype(np.float32), y


def test_dbscan_matches_sklearn():
    X, _ = _data()
    model = DBSCAN(eps=1.0, min_samples=5).fit(DataFrame.from_numpy(X))
    out = model.transform(DataFrame.from_numpy(X))
    labels = np.asarray(out["prediction"])
    sk = SkDBSCAN(eps=1.0, min_samples=5).fit(X)
    assert adjusted_rand_score(labels, sk.labels_) == 1.0
    # same noise set
    assert np.array_equal(labels == -1, sk.labels_ == -1)


def test_dbscan_noise_detection():
    X, _ = _data(n=200)
    X = np.vstack([X, np.full((3, 4), 50.0, dtype=np.float32)])  # outliers
    model = DBSCAN(eps=1.0, min_samples=5).fit(DataFrame.from_numpy(X))
    out = model.transform(DataFrame.from_numpy(X))
    labels = np.asarray(out["prediction"])
    assert (labels[-3:] == -1).all()


def _dist_dbscan(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X, _ = _data(n=300, seed=seed)
    shard = X[comm.rank :: comm.world_size]
    model = DBSCAN(eps=1.0, min_samples=5).fit(DataFrame.from_numpy(shard))
    out = model.transform(DataFrame.from_numpy(shard))
    return np.asarray(out["prediction"])


def test_dbscan_distributed_matches_sklearn():
    results = run_distributed(_dist_dbscan, world_size=2, args=(0,))
    X, _ = _data(n=300, seed=0)
    sk = SkDBSCAN(eps=1.0, min_samples=5).fit(X)
    # reassemble interleaved shards
    labels = np.empty(len(X))
    labels[0::2] = results[0]
    labels[1::2] = results[1]
    assert adjusted_rand_score(labels, sk.labels_) == 1.0


class _FakeSweepExt:
    """torch reference semantics of the dbscan_sweep kernel, honoring the
    pruned tile list (shared by the single- and multi-rank fake tests)."""

    @staticmethod
    def dbscan_sweep(Xf, x_sq, row0, n_rows, eps2, mode, core, labels,
                     tile_idx, tile_off):
        import torch

        BIG = torch.iinfo(torch.int32).max
        n = Xf.shape[0]
        d2 = torch.cdist(Xf[row0 : row0 + n_rows], Xf) ** 2
        if tile_off.numel():
            allowed = torch.zeros((n_rows, n), dtype=torch.bool)
            for b in range((n_rows + 127) // 128):
                for c in tile_idx[tile_off[b] : tile_off[b + 1]].tolist():
                    allowed[b * 128 : (b + 1) * 128, c * 128 : (c + 1) * 128] = True
            d2 = torch.where(allowed, d2, torch.full_like(d2, float("inf")))
        if mode == 0:
            return (d2 <= eps2).sum(dim=1).to(torch.int32)
        masked = torch.where(
            (core[None, :] > 0) & (d2 <= eps2),
            labels[None, :].expand(n_rows, Xf.shape[0]),
            torch.full((1,), BIG, dtype=torch.int32),
        )
        return masked.min(dim=1).values


def _dist_dbscan_rbc(seed: int):
    """Exercise the rbc path's DISTRIBUTED logic (perm broadcast from rank
    0, balanced permuted slices, allgather + unpermute) on 2 gloo ranks with
    the fake kernel."""
    import torch

    import spark_rapids_ml_amd.ops.dispatch as dispatch
    from spark_rapids_ml_amd.parallel.context import get_comm

    dispatch.hip_ops = lambda: _FakeSweepExt
    comm = get_comm()
    # n must clear the rbc n>=4096 gate in _cluster_hip
    X, _ = _data(n=5000, seed=seed)
    n = len(X)
    base, rem = divmod(n, comm.world_size)
    n_local = base + (1 if comm.rank < rem else 0)
    off = comm.rank * base + min(comm.rank, rem)
    Xf = torch.from_numpy(X.astype(np.float32))  # replicated, original order
    model = DBSCAN(eps=1.0, min_samples=5, algorithm="rbc").fit(
        DataFrame.from_numpy(X[off : off + n_local])
    )
    return model._cluster_hip(
        Xf, (Xf * Xf).sum(dim=1), off, n_local, comm, 1.0, 5
    )


def test_dbscan_distributed_rbc_matches_sklearn():
    results = run_distributed(_dist_dbscan_rbc, world_size=2, args=(7,))
    X, _ = _data(n=5000, seed=7)
    labels = np.concatenate(results)  # contiguous rank slices, original order
    sk = SkDBSCAN(eps=1.0, min_samples=5).fit(X)
    assert adjusted_rand_score(labels, sk.labels_) == 1.0
    assert np.array_equal(labels == -1, sk.labels_ == -1)


def test_dbscan_model_persistence(tmp_model_path):
    from spark_rapids_ml_amd.models.clustering import DBSCANModel

    model = DBSCAN(eps=0.8, min_samples=7).fit(DataFrame.from_numpy(_data()[0]))
    model.save(tmp_model_path)
    loaded = DBSCANModel.load(tmp_model_path)
    assert loaded.getEps() == 0.8
    assert loaded.getMinSamples() == 7
    X, _y = _data()
    out = loaded.transform(DataFrame.from_numpy(X))
    assert "prediction" in out.columns


def test_cluster_hip_logic_with_fake_kernel(monkeypatch):
    """The GPU clustering path's sweep-loop/fixpoint/border logic, exercised
    on CPU by faking the dbscan_sweep kernel with its torch reference
    semantics (the real kernel's numerics are tested on GPU in
    test_hip_ops)."""
    import torch

    from spark_rapids_ml_amd.models import clustering as mod

    BIG = torch.iinfo(torch.int32).max

    class FakeExt:
        @staticmethod
        def dbscan_sweep(Xf, x_sq, row0, n_rows, eps2, mode, core, labels,
                         tile_idx, tile_off):
            n = Xf.shape[0]
            d2 = torch.cdist(Xf[row0 : row0 + n_rows], Xf) ** 2
            if tile_off.numel():
                # honor the pruned tile list: columns outside the admissible
                # tiles are invisible, exactly as in the HIP kernel — wrong
                # lists would therefore produce wrong labels below
                allowed = torch.zeros((n_rows, n), dtype=torch.bool)
                for b in range((n_rows + 127) // 128):
                    for c in tile_idx[tile_off[b] : tile_off[b + 1]].tolist():
                        allowed[b * 128 : (b + 1) * 128, c * 128 : (c + 1) * 128] = True
                d2 = torch.where(allowed, d2, torch.full_like(d2, float("inf")))
            if mode == 0:
                return (d2 <= eps2).sum(dim=1).to(torch.int32)
            masked = torch.where(
                (core[None, :] > 0) & (d2 <= eps2),
                labels[None, :].expand(n_rows, Xf.shape[0]),
                torch.full((1,), BIG, dtype=torch.int32),
            )
            return masked.min(dim=1).values

    import spark_rapids_ml_amd.ops.dispatch as dispatch

    monkeypatch.setattr(dispatch, "hip_ops", lambda: FakeExt)

    X, y = _data(n=400)
    from sklearn.cluster import DBSCAN as SkDBSCAN
    from sklearn.metrics import adjusted_rand_score

    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    Xf = torch.from_numpy(X.astype(np.float32))
    sk = SkDBSCAN(eps=0.6, min_samples=5).fit(X)
    for algo in ("brute", "rbc"):
        model = DBSCAN(eps=0.6, min_samples=5, algorithm=algo).fit(
            DataFrame.from_numpy(X)
        )
        # call the hip path directly (CPU tensors + fake kernel)
        labels = model._cluster_hip(
            Xf, (Xf * Xf).sum(dim=1), 0, len(X), comm, 0.6 * 0.6, 5
        )
        assert adjusted_rand_score(sk.labels_, labels) > 0.95, algo
        assert np.array_equal(labels == -1, sk.labels_ == -1), algo


def test_dbscan_rbc_tile_lists_conservative():
    """Every pair within eps must land in an admissible (row-block,
    column-tile) pair — the triangle-inequality bound may keep extra tiles
    but must never drop a real neighbor."""
    import torch

    from spark_rapids_ml_amd.models.clustering import (
        _coarse_assign,
        _tile_lists,
        _tile_stats,
    )

    rng = np.random.default_rng(3)
    X, _ = _data(n=2000, seed=3)
    Xf = torch.from_numpy(X.astype(np.float32))
    perm = torch.argsort(_coarse_assign(Xf, 16, iters=2), stable=True)
    Xp = Xf[perm].contiguous()
    eps = 1.0
    tile_idx, tile_off, kept = _tile_lists(Xp, 0, len(X), eps)
    assert 0.0 < kept <= 1.0
    nb = (len(X) + 127) // 128
    assert tile_off.shape[0] == nb + 1
    admissible = set()
    for b in range(nb):
        for c in tile_idx[tile_off[b] : tile_off[b + 1]].tolist():
            admissible.add((b, int(c)))
    d2 = torch.cdist(Xp, Xp) ** 2
    ii, jj = torch.nonzero(d2 <= eps * eps, as_tuple=True)
    for i, j in zip(ii.tolist(), jj.tolist()):
        assert (i // 128, j // 128) in admissible
    # on clustered data the bound actually prunes
    c, r = _tile_stats(Xp)
    assert c.shape == (nb, X.shape[1]) and r.shape == (nb,)
    assert kept < 1.0
