"""GPU numerics: HIP kernels vs plain-torch fp32 references (same op).

Every test is @pytest.mark.gpu and compares the gfx950 kernel output against
ops.torch_ref computed on the same device tensors.
"""

import numpy as np
import pytest
import torch

from spark_rapids_ml_amd.ops import torch_ref

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from spark_rapids_ml_amd.ops.dispatch import hip_ops

    return hip_ops()


def _rand(n, d, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, d, generator=g, dtype=torch.float32).cuda()


@pytest.mark.parametrize("n,d,k", [(1000, 64, 10), (4096, 128, 100), (1025, 37, 33), (513, 3000, 257)])
def test_kmeans_assign_matches_ref(ext, n, d, k):
    X = _rand(n, d)
    C = _rand(k, d, seed=1)
    x_sq = (X * X).sum(dim=1)
    labels, min_d, inertia = ext.kmeans_assign(X, C, x_sq)
    ref_labels, _, _, ref_inertia = torch_ref.kmeans_assign_reduce(X, C, x_sq)
    l1, l2 = labels.cpu().numpy(), ref_labels.cpu().numpy()
    # ties can break either way; distances must agree
    mismatch = l1 != l2
    if mismatch.any():
        d1 = ((X[mismatch] - C[labels[mismatch].long()]) ** 2).sum(1)
        d2 = ((X[mismatch] - C[ref_labels[mismatch].long()]) ** 2).sum(1)
        assert torch.allclose(d1, d2, rtol=1e-4, atol=1e-3)
    assert abs(float(inertia.item()) - ref_inertia) / max(1.0, abs(ref_inertia)) < 1e-4


@pytest.mark.parametrize(
    "n,d,k",
    [
        (4096, 128, 100),    # LDS-privatized scatter path (k*d*4 <= 120KB, d even)
        (1000, 65, 7),       # LDS path, odd d (scalar loads)
        (100000, 128, 200),  # BASELINE km100m shape (LDS path)
        (2048, 3000, 40),    # k*d*4 > 120KB -> sort+segment path
    ],
)
def test_label_accumulate_matches_ref(ext, n, d, k):
    X = _rand(n, d)
    labels = torch.randint(0, k, (n,), dtype=torch.int32).cuda()
    sums, counts = ext.label_accumulate(X, labels, k)
    ref_sums = torch.zeros(k, d, dtype=torch.float64).cuda()
    ref_sums.index_add_(0, labels.long(), X.double())
    ref_counts = torch.bincount(labels.long(), minlength=k).double()
    assert torch.allclose(counts.double(), ref_counts)
    assert torch.allclose(sums.double(), ref_sums, rtol=1e-4, atol=1e-2)


@pytest.mark.parametrize("n,d", [(1000, 64), (4096, 128), (777, 301), (2048, 3000)])
def test_gram_matches_ref(ext, n, d):
    X = _rand(n, d)
    G = ext.gram_f32(X)
    ref = X.T @ X
    assert torch.allclose(G, ref, rtol=1e-4, atol=1e-2), (G - ref).abs().max()
    # symmetric
    assert torch.allclose(G, G.T)


@pytest.mark.parametrize("n,C", [(1000, 1), (1000, 3), (513, 17)])
def test_softmax_residual_matches_ref(ext, n, C):
    g = torch.Generator().manual_seed(0)
    scores = torch.randn(n, C, generator=g, dtype=torch.float32).cuda()
    y = torch.randint(0, max(2, C), (n,), generator=g).cuda()
    if C == 1:
        y = (y % 2).long()
    resid, loss = ext.softmax_residual_loss(scores, y)
    if C == 1:
        z = scores[:, 0]
        t = y.float() * 2 - 1
        ref_loss = torch.nn.functional.softplus(-t * z).sum()
        ref_resid = (torch.sigmoid(z) - y.float()).unsqueeze(1)
    else:
        logp = torch.log_softmax(scores, dim=1)
        ref_loss = -logp.gather(1, y.view(-1, 1)).sum()
        ref_resid = torch.exp(logp)
        ref_resid.scatter_add_(1, y.view(-1, 1), -torch.ones(n, 1).cuda())
    assert torch.allclose(loss, ref_loss, rtol=1e-4, atol=1e-2)
    assert torch.allclose(resid, ref_resid, rtol=1e-3, atol=1e-4)


def test_hip_required_on_gpu():
    """On a GPU box the extension must be present and loaded (no silent
    eager fallback)."""
    from spark_rapids_ml_amd.ops.dispatch import has_hip_ops, use_hip

    assert has_hip_ops()
    X = torch.zeros(4, 4).cuda()
    assert use_hip(X)


@pytest.mark.parametrize("nq,ni,d,k", [(500, 2000, 64, 8), (64, 128, 32, 5), (1000, 10000, 768, 64), (130, 999, 33, 16)])
def test_knn_select_matches_ref(ext, nq, ni, d, k):
    Q = _rand(nq, d, seed=3)
    I = _rand(ni, d, seed=4)
    d2, idx = ext.knn_select(Q, I, k)
    ref_d, ref_i = torch_ref.knn_topk(Q, I, k)
    dists = torch.sqrt(torch.clamp(d2, min=0))
    assert torch.allclose(dists, ref_d, rtol=1e-3, atol=1e-3), (dists - ref_d).abs().max()
    # indices equal up to distance ties
    mism = idx != ref_i
    if bool(mism.any()):
        assert torch.allclose(dists[mism], ref_d[mism], rtol=1e-3, atol=1e-3)


@pytest.mark.parametrize("task,nc", [("classification", 3), ("regression", 0)])
def test_rf_histogram_matches_ref(ext, task, nc):
    g = torch.Generator().manual_seed(0)
    n, d, nb, B, mf = 5000, 40, 16, 7, 6
    Xb = torch.randint(0, nb, (n, d), generator=g, dtype=torch.uint8).cuda()
    loc = torch.randint(0, B, (n,), generator=g).cuda()
    feat_sel = torch.stack([torch.randperm(d, generator=g)[:mf] for _ in range(B)]).cuda()
    sl, perm_rel = loc.sort()
    perm = perm_rel.to(torch.int64).contiguous()
    seg_off = torch.searchsorted(sl, torch.arange(B + 1).cuda()).contiguous()
    if task == "classification":
        y = torch.randint(0, nc, (n,), generator=g, dtype=torch.int32).cuda()
        H = ext.rf_histogram(Xb.T.contiguous(), perm, seg_off, feat_sel.to(torch.int32), y, 0, mf, nb, nc,
                             torch.empty(0, dtype=torch.int32).cuda(), 0.0)
        ref = torch.zeros(B, mf, nb, nc).cuda()
        for b in range(B):
            m = loc == b
            sel = feat_sel[b]
            for q in range(mf):
                bins = Xb[m][:, sel[q]].long()
                for c in range(nc):
                    ref[b, q].index_add_(
                        0, bins[y[m] == c], torch.ones(int((y[m] == c).sum())).cuda()
                    ) if False else None
                ref[b, q] = torch.zeros(nb, nc).cuda().index_put_(
                    (bins, y[m].long()), torch.ones(len(bins)).cuda(), accumulate=True
                )
    else:
        y = torch.randn(n, generator=g).cuda()
        H = ext.rf_histogram(Xb.T.contiguous(), perm, seg_off, feat_sel.to(torch.int32), y, 0, mf, nb, 0,
                             torch.empty(0, dtype=torch.int32).cuda(),
                             float(y.abs().max().item()))
        ref = torch.zeros(B, mf, nb, 2).cuda()
        for b in range(B):
            m = loc == b
            sel = feat_sel[b]
            for q in range(mf):
                bins = Xb[m][:, sel[q]].long()
                ref[b, q, :, 0].index_add_(0, bins, torch.ones(len(bins)).cuda())
                ref[b, q, :, 1].index_add_(0, bins, y[m])
    assert torch.allclose(H, ref, rtol=1e-4, atol=1e-2), (H - ref).abs().max()


@pytest.mark.parametrize("classif,C", [(True, 3), (False, 2)])
def test_rf_best_split_matches_ref(ext, classif, C):
    from spark_rapids_ml_amd.models.tree import _best_split_class, _best_split_reg, _max_lastdim

    g = torch.Generator().manual_seed(1)
    B, F, nb = 9, 23, 32
    H = (torch.rand(B, F, nb, C, generator=g) * 10).round().cuda()
    gain, feat, bins, lval, rval = ext.rf_best_split(H, 2, classif)
    if classif:
        rg, rb, rl, rr, _ = _best_split_class(H, 2)
    else:
        rg, rb, rl, rr, _ = _best_split_reg(H, 2)
    g2, f2 = _max_lastdim(rg)
    valid = g2 > 0
    assert torch.allclose(gain[valid], g2[valid], rtol=1e-4, atol=1e-5), (gain - g2).abs().max()
    # feature/bin may differ on exact gain ties; check gains at chosen splits
    ar = torch.arange(B).cuda()
    same = (feat.long() == f2) & valid
    assert torch.allclose(
        bins[same].long(), rb[ar[same], f2[same]]
    ) or torch.allclose(gain[same], g2[same], rtol=1e-5)
    lk = lval[same]
    if not classif:
        # kernel emits raw (count, sum); torch reference emits (mean, count)
        lk = torch.stack([lk[:, 1] / lk[:, 0].clamp(min=1e-12), lk[:, 0]], dim=1)
    assert torch.allclose(lk, rl[ar[same], f2[same]], rtol=1e-4, atol=1e-3)


@pytest.mark.gpu
def test_dbscan_sweep_matches_torch():
    from spark_rapids_ml_amd.ops.dispatch import hip_ops

    ext = hip_ops()
    dev = torch.device("cuda:0")
    rng = np.random.default_rng(0)
    n, d = 3000, 37  # non-multiple-of-64 d exercises the tail guards
    X = torch.from_numpy(rng.normal(size=(n, d)).astype(np.float32)).to(dev)
    x_sq = (X * X).sum(dim=1)
    eps2 = float(np.quantile(rng.normal(size=1000) ** 2 * d, 0.02))
    d2 = torch.cdist(X, X) ** 2
    # mode 0: neighbor counts
    empty_u8 = torch.empty(0, dtype=torch.uint8, device=dev)
    empty_i32 = torch.empty(0, dtype=torch.int32, device=dev)
    counts = ext.dbscan_sweep(
        X, x_sq, 0, n, eps2, 0, empty_u8, empty_i32, empty_i32, empty_i32
    )
    ref_counts = (d2 <= eps2).sum(dim=1).to(torch.int32)
    mism = (counts != ref_counts).sum().item()
    assert mism <= n * 0.003, f"{mism} count mismatches"  # eps-boundary f32 ties
    # mode 1: min core-neighbor label, on a row slice
    core = (ref_counts >= 5).to(torch.uint8)
    labels = torch.arange(n, dtype=torch.int32, device=dev)
    labels[core == 0] = torch.iinfo(torch.int32).max
    row0, n_rows = 1000, 1500
    got = ext.dbscan_sweep(
        X, x_sq, row0, n_rows, eps2, 1, core, labels, empty_i32, empty_i32
    )
    BIG = torch.iinfo(torch.int32).max
    masked = torch.where(
        (core[None, :] > 0) & (d2[row0 : row0 + n_rows] <= eps2),
        labels[None, :].expand(n_rows, n),
        torch.full((1,), BIG, dtype=torch.int32, device=dev),
    )
    ref = masked.min(dim=1).values
    mism = (got != ref).sum().item()
    assert mism <= n_rows * 0.003, f"{mism} label mismatches"


@pytest.mark.parametrize("n,B", [(100000, 1), (50000, 977), (200000, 4096)])
def test_rf_partition_matches_sort(ext, n, B):
    g = torch.Generator().manual_seed(3)
    n_nodes = B * 3 + 5
    node_of_row = torch.randint(0, n_nodes, (n,), generator=g).cuda()
    batch = torch.randperm(n_nodes, generator=g)[:B].cuda()
    lut = torch.full((n_nodes,), -1, dtype=torch.int64).cuda()
    lut[batch] = torch.arange(B, dtype=torch.int64).cuda()
    perm, seg_off = ext.rf_partition(node_of_row, lut, B)

    # reference: torch sort path
    local = lut[node_of_row]
    rows = torch.nonzero(local >= 0).flatten()
    loc = local[rows]
    sl, perm_rel = loc.sort()
    ref_seg = torch.searchsorted(sl, torch.arange(B + 1, dtype=torch.int64).cuda())
    assert torch.equal(seg_off.cpu(), ref_seg.cpu())
    total = int(seg_off[-1].item())
    assert total == rows.numel()
    # each segment must hold exactly the rows of its node (order-free)
    got = perm[:total]
    assert torch.equal(lut[node_of_row[got]].cpu().sort().values,
                       sl.cpu())
    for b in [0, B // 2, B - 1]:
        s, e = int(seg_off[b]), int(seg_off[b + 1])
        seg_rows = got[s:e]
        assert (lut[node_of_row[seg_rows]] == b).all()
    # perm must be a permutation of the in-batch rows
    assert torch.equal(got.sort().values.cpu(), rows.sort().values.cpu())


def test_rf_reroute_matches_torch(ext):
    g = torch.Generator().manual_seed(5)
    n, d, ns = 100000, 32, 37
    n_nodes = 200
    Xb = torch.randint(0, 64, (n, d), generator=g, dtype=torch.uint8).cuda()
    node_of_row = torch.randint(0, n_nodes, (n,), generator=g).cuda()
    split_nodes = torch.randperm(n_nodes, generator=g)[:ns].cuda()
    lut2 = torch.full((n_nodes,), -1, dtype=torch.int64).cuda()
    lut2[split_nodes] = torch.arange(ns, dtype=torch.int64).cuda()
    f_t = torch.randint(0, d, (ns,), generator=g, dtype=torch.int32).cuda()
    b_t = torch.randint(0, 64, (ns,), generator=g, dtype=torch.int32).cuda()
    l_t = torch.arange(n_nodes, n_nodes + ns, dtype=torch.int64).cuda()
    r_t = l_t + ns

    ref = node_of_row.clone()
    sl = lut2[ref]
    mrows = torch.nonzero(sl >= 0).flatten()
    srel = sl[mrows]
    go_left = Xb[mrows, f_t[srel].long()].to(torch.int64) <= b_t[srel].long()
    ref[mrows] = torch.where(go_left, l_t[srel], r_t[srel])

    got = node_of_row.clone()
    ext.rf_reroute(got, lut2, f_t, b_t, l_t, r_t, Xb.T.contiguous(),
                   torch.empty(0, dtype=torch.int32).cuda())
    assert torch.equal(got.cpu(), ref.cpu())


@pytest.mark.parametrize("nq,ni,d,k", [(500, 20000, 64, 64), (333, 7777, 100, 17), (64, 500, 768, 64)])
def test_knn_gemm_select_matches_ref(ext, nq, ni, d, k):
    from spark_rapids_ml_amd.ops.knn import _knn_topk_gemm_select

    Q = _rand(nq, d, seed=7)
    I = _rand(ni, d, seed=8)
    dd, ii = _knn_topk_gemm_select(Q, I, k, chunk_elems=1 << 22)  # force chunking
    rd, ri = torch_ref.knn_topk(Q, I, k)
    assert torch.allclose(dd.cpu(), rd.cpu(), rtol=1e-3, atol=1e-2)
    # indices can differ on ties; distances of chosen indices must match
    same = (ii.cpu() == ri.cpu()).float().mean().item()
    assert same > 0.99


@pytest.mark.parametrize("r,n,d,m", [(10000, 10000, 256, 40), (5000, 7000, 100, 16), (1000, 1000, 2048, 8)])
def test_gather_dists_matches_torch(ext, r, n, d, m):
    g = torch.Generator().manual_seed(9)
    A = _rand(r, d, seed=10)
    B = _rand(n, d, seed=11)
    cand = torch.randint(0, n, (r, m), generator=g).cuda()
    got = ext.gather_dists(A, B, cand)
    a = A[:, None, :]
    b = B[cand]
    ref = ((a - b) ** 2).sum(dim=2)
    assert torch.allclose(got, ref, rtol=1e-3, atol=1e-2)


def test_rf_histogram_virtual_rows(ext):
    """Histogram over a VIRTUAL row space (forest arena): sample maps
    virtual->physical; with sample empty, phys = v %% n_phys."""
    g = torch.Generator().manual_seed(11)
    n, d, nb, nc, T = 5000, 8, 16, 3, 4
    Xb = torch.randint(0, nb, (n, d), generator=g, dtype=torch.uint8).cuda()
    y = torch.randint(0, nc, (n,), generator=g, dtype=torch.int32).cuda()
    vn = T * n
    sample = torch.randint(0, n, (vn,), generator=g, dtype=torch.int32).cuda()
    node_of_row = torch.arange(vn).cuda() // n  # node t = tree t root
    lut = torch.arange(T, dtype=torch.int64).cuda()
    perm, seg_off = ext.rf_partition(node_of_row, lut, T)
    feat_sel = torch.empty((0, 0), dtype=torch.int32).cuda()
    H = ext.rf_histogram(Xb.T.contiguous(), perm, seg_off, feat_sel, y, 0, d, nb, nc, sample, 0.0)
    # reference per tree
    for t in range(T):
        rows = sample[t * n : (t + 1) * n].long()
        for f in [0, d - 1]:
            ref = torch.zeros(nb, nc).cuda()
            idx = Xb[rows, f].long() * nc + y[rows].long()
            ref.view(-1).scatter_add_(0, idx, torch.ones_like(idx, dtype=torch.float32))
            assert torch.allclose(H[t, f], ref), (t, f)


@pytest.mark.parametrize("task", ["classification", "regression"])
def test_fil_predict_matches_torch_traversal(ext, task):
    """fil_predict kernel vs the level-wise torch traversal on a real fit."""
    import numpy as np

    from sklearn.datasets import make_classification, make_regression

    from spark_rapids_ml_amd import RandomForestClassifier, RandomForestRegressor
    from spark_rapids_ml_amd.data import DataFrame

    if task == "classification":
        X, y = make_classification(n_samples=3000, n_features=12, n_classes=3,
                                   n_informative=6, n_clusters_per_class=1,
                                   random_state=0)
        est = RandomForestClassifier(numTrees=7, maxDepth=6, seed=1)
    else:
        X, y = make_regression(n_samples=3000, n_features=12, noise=3.0, random_state=0)
        est = RandomForestRegressor(numTrees=7, maxDepth=6, seed=1)
    X = X.astype(np.float32)
    m = est.fit(DataFrame.from_numpy(X, y.astype(np.float64)))

    Xt = torch.from_numpy(X).cuda()
    got = m._predict_raw(X)  # GPU -> fil kernel path

    # torch reference traversal on device
    device = Xt.device
    n = Xt.shape[0]
    if task == "classification":
        ref = torch.zeros((n, m.numClasses), dtype=torch.float32, device=device)
    else:
        ref = torch.zeros(n, dtype=torch.float32, device=device)
    for t in m.trees:
        feature = torch.from_numpy(t["feature"]).to(device, torch.int64)
        thr = torch.from_numpy(t["threshold"]).to(device)
        left = torch.from_numpy(t["left"]).to(device, torch.int64)
        right = torch.from_numpy(t["right"]).to(device, torch.int64)
        leaf = torch.from_numpy(t["is_leaf"]).to(device)
        value = torch.from_numpy(t["value"]).to(device)
        node = torch.zeros(n, dtype=torch.int64, device=device)
        while True:
            at_leaf = leaf[node]
            if bool(at_leaf.all()):
                break
            f = feature[node].clamp(min=0)
            xv = Xt.gather(1, f.view(-1, 1)).flatten()
            go_left = xv < thr[node]
            nxt = torch.where(go_left, left[node], right[node])
            node = torch.where(at_leaf, node, nxt)
        v = value[node]
        if task == "classification":
            ref += v / torch.clamp(v.sum(dim=1, keepdim=True), min=1e-12)
        else:
            ref += v[:, 0]
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-4)


def test_csr_glm_kernels_match_torch(ext):
    """csr_fwd / csr_grad / csr_col_moments vs torch sparse reference."""
    import numpy as np
    import scipy.sparse as sp

    rng = np.random.default_rng(0)
    n, d, C = 50000, 512, 3
    Xc = sp.random(n, d, density=0.02, format="csr", dtype=np.float64,
                   random_state=np.random.RandomState(1)).astype(np.float32)
    indptr = torch.from_numpy(Xc.indptr.astype(np.int32)).cuda()
    indices = torch.from_numpy(Xc.indices.astype(np.int32)).cuda()
    vals = torch.from_numpy(Xc.data).cuda()
    W = torch.randn(C, d, generator=torch.Generator().manual_seed(2)).cuda()

    scores = ext.csr_fwd(indptr, indices, vals, W.T.contiguous())
    Xt = torch.sparse_csr_tensor(indptr, indices, vals, size=Xc.shape)
    ref_scores = torch.sparse.mm(Xt, W.T.contiguous())
    assert torch.allclose(scores, ref_scores, rtol=1e-4, atol=1e-4)

    resid = torch.randn(n, C, generator=torch.Generator().manual_seed(3)).cuda()
    grad = ext.csr_grad(indptr, indices, vals, resid.contiguous(), d)
    ref_grad = torch.sparse.mm(Xt.t().to_sparse_csr(), resid).T
    assert torch.allclose(grad, ref_grad, rtol=1e-3, atol=1e-2)

    mom = ext.csr_col_moments(indices, vals, d)
    dense = torch.from_numpy(np.asarray(Xc.todense(), np.float64)).cuda()
    assert torch.allclose(mom[0], dense.sum(0), rtol=1e-6, atol=1e-6)
    assert torch.allclose(mom[1], (dense * dense).sum(0), rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("n,d,k", [(50000, 128, 100), (12345, 64, 37), (8000, 3000, 1000)])
def test_kmeans_gemm_argmin_matches_fused(ext, n, d, k):
    """GEMM + kmeans_argmin_kn epilogue vs the all-in-one assign kernel."""
    X = _rand(n, d, seed=21)
    C = _rand(k, d, seed=22)
    x_sq = (X * X).sum(dim=1)
    from spark_rapids_ml_amd.ops.kmeans import _assign_gemm

    labels_g, inertia_g = _assign_gemm(ext, X, C, x_sq, n, k,
                                       max_dots_bytes=16 << 20)  # force chunking
    labels_f, _md, inertia_f = ext.kmeans_assign(X, C, x_sq)
    mism = (labels_g != labels_f)
    if bool(mism.any()):
        # ties may break differently; winning distances must agree
        d_g = ((X[mism] - C[labels_g[mism].long()]) ** 2).sum(1)
        d_f = ((X[mism] - C[labels_f[mism].long()]) ** 2).sum(1)
        assert torch.allclose(d_g, d_f, rtol=1e-3, atol=1e-2)
    assert abs(float(inertia_g.item()) - float(inertia_f.item())) <= (
        1e-4 * max(1.0, abs(float(inertia_f.item())))
    )


@pytest.mark.parametrize("n,d,k", [(50000, 128, 200), (12345, 64, 37), (7003, 37, 5)])
def test_kmeans_gemm_argmin_nk_matches_fused(ext, n, d, k):
    """Tall-skinny GEMM + kmeans_argmin_nk (the small-k path) vs the
    all-in-one assign kernel."""
    X = _rand(n, d, seed=31)
    C = _rand(k, d, seed=32)
    x_sq = (X * X).sum(dim=1)
    from spark_rapids_ml_amd.ops.kmeans import _assign_gemm_nk

    labels_g, inertia_g = _assign_gemm_nk(ext, X, C, x_sq, n, k,
                                          max_dots_bytes=4 << 20)  # force chunking
    labels_f, _md, inertia_f = ext.kmeans_assign(X, C, x_sq)
    mism = (labels_g != labels_f)
    if bool(mism.any()):
        d_g = ((X[mism] - C[labels_g[mism].long()]) ** 2).sum(1)
        d_f = ((X[mism] - C[labels_f[mism].long()]) ** 2).sum(1)
        assert torch.allclose(d_g, d_f, rtol=1e-3, atol=1e-2)
    assert abs(float(inertia_g.item()) - float(inertia_f.item())) <= (
        1e-4 * max(1.0, abs(float(inertia_f.item())))
    )
