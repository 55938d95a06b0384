"""UMAP embedding quality (pattern: reference tests/test_umap.py uses
trustworthiness)."""

import numpy as np
import pytest
from sklearn.datasets import make_blobs
from sklearn.manifold import trustworthiness

from spark_rapids_ml_amd import UMAP, UMAPModel
from spark_rapids_ml_amd.data import DataFrame


def _data(n=400, d=10, seed=0):
    X, y = make_blobs(n_samples=n, n_features=d, centers=5, cluster_std=0.5, random_state=seed)
    return X.astype(np.float32), y


def test_umap_trustworthiness():
    X, y = _data()
    model = UMAP(n_neighbors=15, n_epochs=200, random_state=42).fit(DataFrame.from_numpy(X))
    emb = model.embedding
    assert emb.shape == (len(X), 2)
    t = trustworthiness(X, emb, n_neighbors=15)
    assert t > 0.8, f"trustworthiness {t}"


def test_umap_transform_places_near_train():
    X, y = _data(n=300)
    model = UMAP(n_neighbors=10, n_epochs=150, random_state=1).fit(DataFrame.from_numpy(X))
    out = model.transform(DataFrame.from_numpy(X[:50]))
    emb_new = np.asarray(out["embedding"])
    assert emb_new.shape == (50, 2)
    # transformed points should land near their own training embedding
    d = np.linalg.norm(emb_new - model.embedding[:50], axis=1)
    spread = np.linalg.norm(model.embedding.max(0) - model.embedding.min(0))
    assert np.median(d) < spread * 0.25


def test_umap_sample_fraction():
    X, y = _data(n=400)
    model = UMAP(n_neighbors=10, n_epochs=100, sample_fraction=0.5, random_state=3).fit(
        DataFrame.from_numpy(X)
    )
    assert model.rawData.shape[0] == 200


def test_umap_persistence(tmp_model_path):
    X, _ = _data(n=150)
    model = UMAP(n_neighbors=8, n_epochs=80, random_state=5).fit(DataFrame.from_numpy(X))
    model.save(tmp_model_path)
    loaded = UMAPModel.load(tmp_model_path)
    assert np.allclose(loaded.embedding, model.embedding)
    assert np.allclose(loaded.rawData, model.rawData)


def test_find_ab_params():
    from spark_rapids_ml_amd.models.umap import find_ab_params

    a, b = find_ab_params(1.0, 0.1)
    # umap-learn reference values for spread=1, min_dist=0.1
    assert abs(a - 1.577) < 0.05
    assert abs(b - 0.895) < 0.05


def test_umap_model_persistence(tmp_model_path):
    from spark_rapids_ml_amd.models.umap import UMAPModel

    X, _ = _data(n=300)
    model = UMAP(n_neighbors=10, n_epochs=20, random_state=1).fit(DataFrame.from_numpy(X))
    model.save(tmp_model_path)
    loaded = UMAPModel.load(tmp_model_path)
    assert np.allclose(loaded.embedding, model.embedding)
    out = loaded.transform(DataFrame.from_numpy(X[:40]))
    assert np.asarray(out["embedding"]).shape == (40, 2)


def _dist_umap(_):
    from spark_rapids_ml_amd.parallel.context import get_comm
    from tests.test_umap import _data

    comm = get_comm()
    X, _y = _data(n=400)
    shard = X[comm.rank :: comm.world_size]
    model = UMAP(n_neighbors=10, n_epochs=30, random_state=1).fit(DataFrame.from_numpy(shard))
    out = model.transform(DataFrame.from_numpy(shard[:20]))
    return model.embedding, np.asarray(out["embedding"])


def test_umap_distributed_model_replicated():
    from tests.dist_utils import run_distributed

    results = run_distributed(_dist_umap, world_size=2, args=(None,))
    emb0, out0 = results[0]
    emb1, out1 = results[1]
    # fit on gathered data, broadcast: both ranks hold the identical model
    assert np.allclose(emb0, emb1)
    assert out0.shape == (20, 2) and out1.shape == (20, 2)


def test_umap_supervised_separates_classes():
    """Supervised fit (labelCol): cross-label edges shrink by exp(-5), so
    same-label points embed closer than different-label ones (reference
    supervised fit, umap.py:1035-1050)."""
    X, y = _data(n=500)
    df = DataFrame({"features": X, "label": y.astype(np.float64)})
    sup = UMAP(n_neighbors=10, n_epochs=100, random_state=1, labelCol="label").fit(df)
    emb = sup.embedding
    # mean intra-class distance well below mean inter-class distance
    intra, inter = [], []
    rng = np.random.default_rng(0)
    idx = rng.choice(len(emb), size=(400, 2))
    for i, j in idx:
        d = np.linalg.norm(emb[i] - emb[j])
        (intra if y[i] == y[j] else inter).append(d)
    assert np.mean(intra) < 0.5 * np.mean(inter)


def test_fuzzy_simplicial_set_torch_matches_numpy():
    """The device-resident fuzzy-set path must reproduce the numpy/scipy
    reference (same smooth-knn bisection, same symmetrization algebra)."""
    import torch

    from spark_rapids_ml_amd.models.umap import (
        _fuzzy_simplicial_set,
        _fuzzy_simplicial_set_t,
    )

    rng = np.random.default_rng(0)
    n, k = 500, 15
    knn_d = np.sort(np.abs(rng.normal(size=(n, k))), axis=1).astype(np.float64)
    knn_i = np.stack([rng.permutation(n)[:k] for _ in range(n)]).astype(np.int64)

    r1, c1, v1 = _fuzzy_simplicial_set(knn_d.astype(np.float32), knn_i, 1.0, 1.0)
    r2, c2, v2 = _fuzzy_simplicial_set_t(
        torch.from_numpy(knn_d.astype(np.float32)), torch.from_numpy(knn_i), 1.0, 1.0
    )
    # compare as sparse maps (orderings differ)
    m1 = {(int(r), int(c)): float(v) for r, c, v in zip(r1, c1, v1)}
    m2 = {(int(r), int(c)): float(v) for r, c, v in zip(
        r2.numpy(), c2.numpy(), v2.numpy())}
    assert set(m1) == set(m2)
    for key in m1:
        assert abs(m1[key] - m2[key]) < 1e-5, key


def test_spectral_init_lobpcg_matches_scipy_subspace():
    """LOBPCG spectral init must span the same eigen-subspace as the scipy
    path on a small graph (sign/rotation-invariant comparison)."""
    import scipy.sparse as sp
    import torch
    from scipy.sparse.linalg import eigsh

    from spark_rapids_ml_amd.models.umap import _spectral_init_lobpcg

    rng = np.random.default_rng(0)
    n, k = 600, 10
    # symmetric kNN-ish graph
    rows = np.repeat(np.arange(n), k)
    cols = rng.integers(0, n, n * k)
    vals = rng.random(n * k)
    W = sp.coo_matrix((vals, (rows, cols)), shape=(n, n))
    W = (W + W.T).tocoo()

    emb = _spectral_init_lobpcg(
        torch.from_numpy(W.row.astype(np.int64)),
        torch.from_numpy(W.col.astype(np.int64)),
        torch.from_numpy(W.data),
        n, 2, seed=1, iters=200,
    )
    assert emb is not None and emb.shape == (n, 2)

    deg = np.asarray(W.sum(axis=1)).flatten()
    dinv = 1.0 / np.sqrt(np.maximum(deg, 1e-12))
    L = sp.eye(n) - sp.diags(dinv) @ W.tocsr() @ sp.diags(dinv)
    w, v = eigsh(L, k=3, sigma=0.0, which="LM")
    ref = v[:, 1:3]

    # subspace alignment: projection of emb onto ref explains ~all variance
    E = emb.numpy()
    E = E / np.linalg.norm(E, axis=0, keepdims=True)
    R = ref / np.linalg.norm(ref, axis=0, keepdims=True)
    proj = R @ (R.T @ E)
    frac = (proj * E).sum() / 2.0
    assert frac > 0.97, frac
