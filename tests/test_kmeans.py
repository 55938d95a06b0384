"""KMeans correctness vs sklearn (pattern: reference tests/test_kmeans.py —
generate blobs, fit, compare centers within tolerance)."""

import numpy as np
import pytest
from sklearn.datasets import make_blobs

from spark_rapids_ml_amd import KMeans, KMeansModel
from spark_rapids_ml_amd.data import DataFrame

from .dist_utils import run_distributed


def _sorted_centers(C):
    C = np.asarray(C)
    order = np.lexsort(C.T[::-1])
    return C[order]


def _make(n=1000, d=8, k=4, seed=0):
    X, y = make_blobs(n_samples=n, n_features=d, centers=k, cluster_std=0.5, random_state=seed)
    return X.astype(np.float32), y


def test_kmeans_matches_sklearn_blobs():
    from sklearn.cluster import KMeans as SkKMeans

    X, _ = _make()
    df = DataFrame.from_numpy(X)
    model = KMeans(k=4, maxIter=50, seed=3, tol=1e-6).fit(df)
    sk = SkKMeans(n_clusters=4, n_init=10, random_state=0).fit(X)
    ours = _sorted_centers(model.cluster_centers_)
    theirs = _sorted_centers(sk.cluster_centers_)
    assert np.allclose(ours, theirs, atol=0.05), f"{ours}\n!=\n{theirs}"
    # inertia parity
    assert model.trainingCost <= sk.inertia_ * 1.05


def test_kmeans_random_init():
    X, _ = _make()
    df = DataFrame.from_numpy(X)
    model = KMeans(k=4, maxIter=100, seed=5, initMode="random").fit(df)
    assert model.cluster_centers_.shape == (4, 8)
    assert model.trainingCost > 0


def test_kmeans_transform_labels_consistent():
    X, _ = _make()
    df = DataFrame.from_numpy(X)
    model = KMeans(k=4, maxIter=30, seed=1).fit(df)
    out = model.transform(df)
    labels = np.asarray(out["prediction"])
    # assignment must equal nearest-center assignment
    d = ((X[:, None, :] - model.cluster_centers_[None]) ** 2).sum(axis=2)
    assert np.array_equal(labels, d.argmin(axis=1).astype(labels.dtype))


def test_kmeans_predict_single_vector():
    X, _ = _make()
    model = KMeans(k=4, maxIter=20, seed=1).fit(DataFrame.from_numpy(X))
    lab = model.predict(X[0])
    assert 0 <= lab < 4


def test_kmeans_persistence(tmp_model_path):
    X, _ = _make(n=200)
    model = KMeans(k=3, maxIter=10, seed=2).fit(DataFrame.from_numpy(X))
    model.save(tmp_model_path)
    loaded = KMeansModel.load(tmp_model_path)
    assert np.allclose(loaded.cluster_centers_, model.cluster_centers_)
    assert loaded.getOrDefault("k") == 3
    out = loaded.transform(DataFrame.from_numpy(X))
    assert "prediction" in out.columns


def test_kmeans_estimator_persistence(tmp_model_path):
    est = KMeans(k=5, maxIter=7)
    est.save(tmp_model_path)
    loaded = KMeans.load(tmp_model_path)
    assert loaded.getOrDefault("k") == 5
    assert loaded.native_params["max_iter"] == 7


def test_kmeans_k_greater_than_rows_raises():
    X = np.random.rand(3, 2).astype(np.float32)
    with pytest.raises(ValueError):
        KMeans(k=10, maxIter=5).fit(DataFrame.from_numpy(X))


# -- distributed (gloo world_size=2) ----------------------------------------


def _dist_kmeans_fit(seed: int):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X, _ = _make(n=1000, seed=seed)
    shard = X[comm.rank :: comm.world_size]
    model = KMeans(k=4, maxIter=50, seed=3, tol=1e-6).fit(DataFrame.from_numpy(shard))
    return model.cluster_centers_, model.trainingCost


def test_kmeans_distributed_matches_single():
    results = run_distributed(_dist_kmeans_fit, world_size=2, args=(0,))
    c0, cost0 = results[0]
    c1, cost1 = results[1]
    assert np.allclose(c0, c1)  # replicated model identical across ranks
    X, _ = _make(n=1000, seed=0)
    single = KMeans(k=4, maxIter=50, seed=3, tol=1e-6).fit(DataFrame.from_numpy(X))
    assert np.allclose(
        _sorted_centers(c0), _sorted_centers(single.cluster_centers_), atol=0.05
    )


def test_kmeans_rejects_noneuclidean():
    with pytest.raises(ValueError):
        KMeans(distanceMeasure="cosine")


def test_kmeans_compute_cost():
    X, _ = _make(n=300)
    df = DataFrame.from_numpy(X)
    model = KMeans(k=4, maxIter=20, seed=1, tol=1e-6).fit(df)
    cost = model.computeCost(df)
    assert np.isclose(cost, model.trainingCost, rtol=0.05)


def test_kmeans_summary():
    X, _ = _make(n=300)
    model = KMeans(k=4, maxIter=10, seed=1).fit(DataFrame.from_numpy(X))
    s = model.summary
    assert s.numIter >= 1
    assert sum(s.clusterSizes) == 300
    assert s.trainingCost == model.trainingCost


def test_kmeans_degenerate_inputs():
    # fewer distinct points than k: empty clusters allowed, centers finite
    X = np.repeat(np.array([[0.0, 0.0], [5.0, 5.0], [9.0, 1.0]], dtype=np.float32), 20, axis=0)
    m = KMeans(k=5, maxIter=10, seed=1).fit(DataFrame.from_numpy(X))
    assert np.isfinite(m.cluster_centers_).all()
    assert sum(m.summary.clusterSizes) == 60
    # all-identical points
    m2 = KMeans(k=2, maxIter=5, seed=1).fit(DataFrame.from_numpy(np.zeros((50, 3), np.float32)))
    assert np.isfinite(m2.cluster_centers_).all()


def test_model_copy_with_extra_params():
    X, _ = _make(n=200)
    m = KMeans(k=3, maxIter=5, seed=1).fit(DataFrame.from_numpy(X))
    m2 = m.copy({m.getParam("predictionCol"): "p2"})
    assert m2.getOrDefault("predictionCol") == "p2"
    assert m.getOrDefault("predictionCol") == "prediction"  # original untouched
    assert np.allclose(m2.cluster_centers_, m.cluster_centers_)
    assert "p2" in m2.transform(DataFrame.from_numpy(X)).columns


def test_assign_dispatch_gates(monkeypatch):
    """The k-gated assignment dispatch (measured on MI355X, ops/kmeans.py):
    [k,n] GEMM + argmin_kn for 384 <= k (fits LDS), tall-skinny GEMM +
    argmin_nk for k < 384, fused kernel only via SRML_KMEANS_VARIANT."""
    import torch

    import spark_rapids_ml_amd.ops.kmeans as km

    calls = []

    class FakeExt:
        @staticmethod
        def kmeans_assign(X, C, x_sq):
            calls.append("fused")
            lab, _s, _c, inertia = km.torch_ref.kmeans_assign_reduce(X, C, x_sq)
            return lab.to(torch.int32), torch.zeros_like(x_sq), torch.tensor([inertia])

        @staticmethod
        def kmeans_argmin_kn(dots, x_sq, c_sq):
            calls.append("kn")
            d2 = x_sq[None, :] + c_sq[:, None] - 2.0 * dots
            md, lab = d2.min(dim=0)
            return lab.to(torch.int32), md, md.clamp(min=0).double().sum().reshape(1)

        @staticmethod
        def kmeans_argmin_nk(dots, x_sq, c_sq):
            calls.append("nk")
            d2 = x_sq[:, None] + c_sq[None, :] - 2.0 * dots
            md, lab = d2.min(dim=1)
            return lab.to(torch.int32), md, md.clamp(min=0).double().sum().reshape(1)

        @staticmethod
        def label_accumulate(X, labels, k):
            sums = torch.zeros(k, X.shape[1])
            sums.index_add_(0, labels.long(), X)
            cnt = torch.bincount(labels.long(), minlength=k).float()
            return sums, cnt

    monkeypatch.setattr(km, "hip_ops", lambda: FakeExt)
    monkeypatch.setattr(km, "use_hip", lambda X: True)

    g = torch.Generator().manual_seed(0)
    X = torch.randn(500, 16, generator=g)
    for k, expect in [(8, "nk"), (383, "nk"), (384, "kn"), (512, "kn")]:
        calls.clear()
        C = torch.randn(k, 16, generator=g)
        labels, sums, counts, inertia = km.kmeans_assign_reduce(X, C)
        assert calls[0] == expect, (k, calls)
        ref_lab, ref_sums, ref_cnt, ref_in = km.torch_ref.kmeans_assign_reduce(
            X, C, (X * X).sum(dim=1)
        )
        assert (labels.long() == ref_lab.long()).float().mean() > 0.99
        assert abs(float(inertia) - float(ref_in)) < 1e-3 * max(1.0, abs(float(ref_in)))

    # fused stays selectable
    monkeypatch.setenv("SRML_KMEANS_VARIANT", "fused")
    calls.clear()
    km.kmeans_assign_reduce(X, torch.randn(384, 16, generator=g))
    assert calls[0] == "fused"
