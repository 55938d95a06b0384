"""GPU end-to-end algorithm tests (single MI355X)."""

import numpy as np
import pytest
import torch
from sklearn.datasets import make_blobs, make_classification, make_regression

from spark_rapids_ml_amd import (
    KMeans,
    LinearRegression,
    LogisticRegression,
    NearestNeighbors,
    PCA,
    RandomForestClassifier,
)
from spark_rapids_ml_amd.data import DataFrame

pytestmark = pytest.mark.gpu


def test_kmeans_gpu_matches_sklearn():
    from sklearn.cluster import KMeans as SkKMeans

    X, _ = make_blobs(n_samples=5000, n_features=32, centers=8, cluster_std=0.5, random_state=0)
    X = X.astype(np.float32)
    model = KMeans(k=8, maxIter=50, seed=3, tol=1e-6).fit(DataFrame.from_numpy(X))
    sk = SkKMeans(n_clusters=8, n_init=10, random_state=0).fit(X)
    assert model.trainingCost <= sk.inertia_ * 1.05


def test_pca_gpu_matches_sklearn():
    from sklearn.decomposition import PCA as SkPCA

    rng = np.random.default_rng(0)
    X = (rng.normal(size=(2000, 4)) @ rng.normal(size=(4, 64)) + 0.01 * rng.normal(size=(2000, 64))).astype(np.float32)
    model = PCA(k=3).fit(DataFrame.from_numpy(X))
    sk = SkPCA(n_components=3).fit(X)
    for i in range(3):
        c = np.asarray(model.components_)[i]
        s = sk.components_[i]
        assert min(np.abs(c - s).max(), np.abs(c + s).max()) < 1e-2


def test_linreg_gpu_matches_sklearn():
    from sklearn.linear_model import LinearRegression as SkOLS

    X, y = make_regression(n_samples=5000, n_features=64, noise=1.0, random_state=0)
    model = LinearRegression().fit(DataFrame.from_numpy(X.astype(np.float64), y))
    sk = SkOLS().fit(X, y)
    assert np.allclose(model.coefficients, sk.coef_, atol=1e-4)


def test_logreg_gpu_predictions():
    X, y = make_classification(n_samples=5000, n_features=32, n_informative=16, random_state=0)
    model = LogisticRegression(regParam=0.01, maxIter=100).fit(
        DataFrame.from_numpy(X.astype(np.float32), y.astype(np.float64))
    )
    out = model.transform(DataFrame.from_numpy(X.astype(np.float32)))
    assert (np.asarray(out["prediction"]) == y).mean() > 0.72  # sklearn gets 0.741 here


def test_rf_gpu_accuracy():
    X, y = make_classification(n_samples=3000, n_features=20, n_informative=10, random_state=0)
    model = RandomForestClassifier(numTrees=20, maxDepth=8, seed=0).fit(
        DataFrame.from_numpy(X.astype(np.float32), y.astype(np.float64))
    )
    out = model.transform(DataFrame.from_numpy(X.astype(np.float32)))
    assert (np.asarray(out["prediction"]) == y).mean() > 0.9


def test_knn_gpu_exact():
    from sklearn.neighbors import NearestNeighbors as SkNN

    rng = np.random.default_rng(0)
    X = rng.normal(size=(3000, 64)).astype(np.float32)
    Q = rng.normal(size=(100, 64)).astype(np.float32)
    model = NearestNeighbors(k=8).fit(DataFrame.from_numpy(X))
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(Q))
    sk = SkNN(n_neighbors=8).fit(X)
    sk_d, sk_i = sk.kneighbors(Q)
    # distances must match; index ties allowed
    assert np.allclose(np.asarray(knn_df["distances"]), sk_d, atol=1e-3)


def test_umap_gpu_trustworthiness():
    from sklearn.manifold import trustworthiness

    from spark_rapids_ml_amd import UMAP

    X, _ = make_blobs(n_samples=2000, n_features=32, centers=6, cluster_std=0.5, random_state=0)
    X = X.astype(np.float32)
    model = UMAP(n_neighbors=15, n_epochs=150, random_state=42).fit(DataFrame.from_numpy(X))
    t = trustworthiness(X, model.embedding, n_neighbors=15)
    assert t > 0.8


def test_dbscan_gpu_matches_sklearn():
    from sklearn.cluster import DBSCAN as SkDBSCAN
    from sklearn.metrics import adjusted_rand_score

    from spark_rapids_ml_amd import DBSCAN

    X, _ = make_blobs(n_samples=3000, n_features=8, centers=5, cluster_std=0.4, random_state=0)
    X = X.astype(np.float32)
    model = DBSCAN(eps=1.5, min_samples=5).fit(DataFrame.from_numpy(X))
    out = model.transform(DataFrame.from_numpy(X))
    sk = SkDBSCAN(eps=1.5, min_samples=5).fit(X)
    assert adjusted_rand_score(np.asarray(out["prediction"]), sk.labels_) == 1.0


def test_dbscan_gpu_rbc_matches_brute():
    """algorithm="rbc" (coarse-permuted ball-cover tile pruning) gives the
    same clustering as the dense sweep — the pruned HIP kernel runs the same
    per-pair eps test over a conservative tile superset. The tolerance
    covers f32 eps-boundary ties only: the rbc run accumulates the MFMA dot
    products in permuted row order, so pairs within float rounding of eps
    can flip (same 0.3% bound as test_dbscan_sweep_matches_torch); the
    pruning logic itself is proven exact on CPU in
    test_cluster_hip_logic_with_fake_kernel."""
    from sklearn.cluster import DBSCAN as SkDBSCAN
    from sklearn.metrics import adjusted_rand_score

    from spark_rapids_ml_amd import DBSCAN

    X, _ = make_blobs(
        n_samples=30000, n_features=16, centers=12, cluster_std=0.5, random_state=1
    )
    X = X.astype(np.float32)
    df = DataFrame.from_numpy(X)
    brute = np.asarray(
        DBSCAN(eps=1.5, min_samples=5, algorithm="brute").fit(df).transform(df)["prediction"]
    )
    rbc = np.asarray(
        DBSCAN(eps=1.5, min_samples=5, algorithm="rbc").fit(df).transform(df)["prediction"]
    )
    assert adjusted_rand_score(brute, rbc) > 0.999
    assert ((brute == -1) != (rbc == -1)).sum() <= len(X) * 0.003
    sk = SkDBSCAN(eps=1.5, min_samples=5).fit(X)
    assert adjusted_rand_score(rbc, sk.labels_) > 0.999


def test_ann_gpu_recall():
    from sklearn.neighbors import NearestNeighbors as SkNN

    from spark_rapids_ml_amd import ApproximateNearestNeighbors

    rng = np.random.default_rng(0)
    X = rng.normal(size=(20000, 64)).astype(np.float32)
    model = ApproximateNearestNeighbors(
        k=10, algorithm="ivfflat", algoParams={"nlist": 64, "nprobe": 32}
    ).fit(DataFrame.from_numpy(X))
    _, _, knn_df = model.kneighbors(DataFrame.from_numpy(X[:200]))
    idx = np.asarray(knn_df["indices"])
    sk = SkNN(n_neighbors=10).fit(X)
    _, sk_idx = sk.kneighbors(X[:200])
    hits = sum(len(set(a.tolist()) & set(b.tolist())) for a, b in zip(idx, sk_idx))
    assert hits / sk_idx.size > 0.8  # nprobe=nlist/2 on unclustered gaussians


def test_sparse_logreg_gpu():
    import scipy.sparse as sp

    rng = np.random.default_rng(0)
    X = rng.normal(size=(20000, 128))
    X[rng.random(X.shape) < 0.9] = 0.0
    w = rng.normal(size=128)
    y = (X @ w > 0).astype(np.float64)
    Xs = sp.csr_matrix(X.astype(np.float32))
    model = LogisticRegression(regParam=1e-4, maxIter=100).fit(DataFrame.from_numpy(Xs, y))
    out = model.transform(DataFrame.from_numpy(Xs))
    assert (np.asarray(out["prediction"]) == y).mean() > 0.95


def test_ann_pq_and_cagra_gpu():
    from sklearn.neighbors import NearestNeighbors as SkNN

    from spark_rapids_ml_amd import ApproximateNearestNeighbors

    rng = np.random.default_rng(1)
    X = rng.normal(size=(20000, 64)).astype(np.float32)
    sk = SkNN(n_neighbors=10).fit(X)
    _, sk_idx = sk.kneighbors(X[:200])
    for algo, params, floor in [
        ("ivfpq", {"nlist": 64, "nprobe": 16, "refine_ratio": 4.0}, 0.6),
        ("cagra", {"graph_degree": 32, "itopk_size": 128, "max_iterations": 10}, 0.6),
    ]:
        model = ApproximateNearestNeighbors(k=10, algorithm=algo, algoParams=params).fit(
            DataFrame.from_numpy(X)
        )
        _, _, knn_df = model.kneighbors(DataFrame.from_numpy(X[:200]))
        idx = np.asarray(knn_df["indices"])
        hits = sum(len(set(a.tolist()) & set(b.tolist())) for a, b in zip(idx, sk_idx))
        assert hits / sk_idx.size > floor, f"{algo}: {hits / sk_idx.size}"


def test_umap_supervised_gpu():
    from spark_rapids_ml_amd import UMAP

    X, y = make_blobs(n_samples=3000, n_features=16, centers=4, cluster_std=2.0, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float32), y.astype(np.float64))
    model = UMAP(n_neighbors=10, n_epochs=100, random_state=0).setLabelCol("label").fit(df)
    emb = model.embedding
    # supervised embedding should separate classes decently even with overlap
    from sklearn.neighbors import KNeighborsClassifier

    acc = KNeighborsClassifier(5).fit(emb, y).score(emb, y)
    assert acc > 0.9


def test_rfr_gpu_r2():
    from spark_rapids_ml_amd import RandomForestRegressor

    X, y = make_regression(n_samples=5000, n_features=20, n_informative=10, noise=5.0, random_state=0)
    X = X.astype(np.float32)
    model = RandomForestRegressor(numTrees=20, maxDepth=8, seed=0).fit(
        DataFrame.from_numpy(X, y.astype(np.float64))
    )
    out = model.transform(DataFrame.from_numpy(X))
    pred = np.asarray(out["prediction"])
    r2 = 1 - ((pred - y) ** 2).sum() / ((y - y.mean()) ** 2).sum()
    assert r2 > 0.8, r2


def test_kmeans_gpu_float64_inputs():
    # float32_inputs=False keeps f64 end to end: GPU falls back to the
    # torch path (the f32 MFMA kernel must not receive f64)
    X, _ = make_blobs(n_samples=2000, n_features=16, centers=4, random_state=0)
    model = KMeans(k=4, maxIter=10, seed=1, float32_inputs=False).fit(
        DataFrame.from_numpy(X.astype(np.float64))
    )
    assert model.cluster_centers_.dtype == np.float64


def test_single_vector_predict_gpu():
    """predict/predictRaw/predictProbability route 1-row inputs through the
    device transform path (HIP kernels must handle n=1)."""
    X, y = make_classification(n_samples=400, n_features=16, random_state=0)
    X = X.astype(np.float32)
    df = DataFrame.from_numpy(X, y.astype(np.float64))
    lg = LogisticRegression(maxIter=30).fit(df)
    v = X[0]
    assert lg.predict(v) in (0.0, 1.0)
    assert abs(lg.predictProbability(v).sum() - 1.0) < 1e-6
    rf = RandomForestClassifier(numTrees=5, maxDepth=4, seed=1).fit(df)
    assert rf.predict(v) in (0.0, 1.0)
    Xb, _ = make_blobs(n_samples=300, n_features=8, centers=3, random_state=0)
    km = KMeans(k=3, maxIter=15, seed=1).fit(DataFrame.from_numpy(Xb.astype(np.float32)))
    assert 0 <= km.predict(Xb[0].astype(np.float32)) < 3


def test_umap_sgd_kernel_quality():
    """The HIP umap_sgd kernel must produce an embedding of the same quality
    as the torch SGD path (stochastic — compared by trustworthiness, the
    reference's own acceptance metric for UMAP)."""
    from sklearn.datasets import make_blobs
    from sklearn.manifold import trustworthiness

    from spark_rapids_ml_amd import UMAP

    X, _ = make_blobs(n_samples=2000, n_features=24, centers=6, cluster_std=0.6, random_state=0)
    X = X.astype(np.float32)
    model = UMAP(n_neighbors=15, n_epochs=150, random_state=7).fit(DataFrame.from_numpy(X))
    tw = trustworthiness(X, model.embedding, n_neighbors=10)
    assert tw > 0.92, tw
    # transform path (move_tail=False, separate tail embedding)
    out = model.transform(DataFrame.from_numpy(X[:200]))
    emb_q = np.asarray(out["embedding"])
    assert np.isfinite(emb_q).all()


@pytest.mark.gpu
def test_streaming_fit_on_gpu(monkeypatch):
    """The pinned-chunk streaming path on device must match in-memory."""
    import numpy as np

    from sklearn.datasets import make_regression

    from spark_rapids_ml_amd import LinearRegression
    from spark_rapids_ml_amd.data import DataFrame

    X, y = make_regression(n_samples=200_000, n_features=64, noise=2.0, random_state=0)
    X = X.astype(np.float32)
    df = DataFrame.from_numpy(X, y)
    monkeypatch.setenv("SRML_STREAM_CAP_BYTES", str(8 << 20))  # 8 MB cap, 51 MB data
    m_stream = LinearRegression().fit(df)
    monkeypatch.delenv("SRML_STREAM_CAP_BYTES")
    m_mem = LinearRegression().fit(df)
    np.testing.assert_allclose(
        np.asarray(m_stream.coefficients), np.asarray(m_mem.coefficients),
        rtol=1e-3, atol=1e-4,
    )
