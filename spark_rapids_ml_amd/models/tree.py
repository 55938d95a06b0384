"""Random forest: histogram-based GPU trainer + ensemble-parallel fit.

Reference design (reference tree.py, 744 lines): embarrassingly parallel —
trees are divided across workers (`_estimators_per_worker`, tree.py:330-341),
each worker fits its sub-forest on its LOCAL partition only (single-GPU cuML
RF, histogram split finding with n_bins≤128), the serialized sub-forests are
allGather'ed and concatenated (tree.py:424-460). The trainer here is a
from-scratch histogram forest in torch (CPU/ROCm; the histogram build is the
HIP-kernel candidate): depth-wise frontier, per-(node,feature,bin) class
counts / moment stats via scatter-add, vectorized gain scan, per-node
feature subsampling.

Forest serialization is an own format (arrays per tree: feature, threshold,
children, leaf values) playing the role of treelite bytes + JSON dump
(reference tree.py:424-460): pickled for the allGather merge, stored in the
model attributes, and dumpable as JSON for Spark-tree translation parity.
"""

from __future__ import annotations

import math
import pickle
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from ..core import Estimator, Model, _FitContext
from ..data import to_device_tensor
from ..params import (
    HasFeaturesCol,
    HasFeaturesCols,
    HasLabelCol,
    HasPredictionCol,
    Param,
    TypeConverters,
    HasWeightCol,
)
from ..utils import as_numpy


class _RandomForestParams(HasFeaturesCol, HasFeaturesCols, HasLabelCol, HasPredictionCol, HasWeightCol):
    numTrees = Param("rf", "numTrees", "number of trees.", TypeConverters.toInt)
    maxDepth = Param("rf", "maxDepth", "max tree depth.", TypeConverters.toInt)
    maxBins = Param("rf", "maxBins", "max histogram bins (<=256).", TypeConverters.toInt)
    minInstancesPerNode = Param(
        "rf", "minInstancesPerNode", "min rows per child.", TypeConverters.toInt
    )
    minInfoGain = Param("rf", "minInfoGain", "min split gain.", TypeConverters.toFloat)
    subsamplingRate = Param("rf", "subsamplingRate", "bootstrap fraction.", TypeConverters.toFloat)
    featureSubsetStrategy = Param(
        "rf",
        "featureSubsetStrategy",
        "auto|all|sqrt|log2|onethird|n|fraction.",
        TypeConverters.toString,
    )
    seed = Param("rf", "seed", "random seed.", TypeConverters.toInt)
    bootstrap = Param("rf", "bootstrap", "bootstrap rows.", TypeConverters.toBoolean)
    impurity = Param("rf", "impurity", "gini|entropy|variance.", TypeConverters.toString)
    leafCol = Param("rf", "leafCol", "leaf index column (unsupported).", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(
            numTrees=20,
            maxDepth=5,
            maxBins=32,
            minInstancesPerNode=1,
            minInfoGain=0.0,
            subsamplingRate=1.0,
            featureSubsetStrategy="auto",
            seed=0,
            bootstrap=True,
        )

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        # reference tree.py:93-135
        return {
            "numTrees": "n_estimators",
            "maxDepth": "max_depth",
            "maxBins": "n_bins",
            "minInstancesPerNode": "min_samples_leaf",
            "minInfoGain": "min_impurity_decrease",
            "subsamplingRate": "max_samples",
            "featureSubsetStrategy": "max_features",
            "seed": "random_state",
            "bootstrap": "bootstrap",
            "impurity": "split_criterion",
            "weightCol": None,
            "leafCol": None,
        }

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        return {
            "n_estimators": 20,
            "max_depth": 5,
            "n_bins": 32,
            "min_samples_leaf": 1,
            "min_impurity_decrease": 0.0,
            "max_samples": 1.0,
            "max_features": "auto",
            "random_state": 0,
            "bootstrap": True,
            "split_criterion": None,
            "max_batch_size": 16384,
            # cuML-signature knobs accepted for ctor parity (inert here)
            "n_streams": 4,
            "min_samples_split": 2,
            "max_leaves": -1,
            "verbose": False,
        }

    def setLeafCol(self, value: str) -> "_RandomForestParams":
        raise ValueError("'leafCol' is not supported (reference tree.py:110 parity).")

    def getNumTrees(self) -> int:
        return self.getOrDefault("numTrees")

    def getMaxDepth(self) -> int:
        return self.getOrDefault("maxDepth")


def _max_lastdim(t: torch.Tensor):
    """(values, indices) over the last dim. torch-CPU's indexed max/argmax is
    ~30x slower than numpy in this build; route CPU through numpy."""
    if t.is_cuda:
        return t.max(dim=-1)
    arr = t.detach().numpy()
    idx = np.argmax(arr, axis=-1)
    vals = np.take_along_axis(arr, np.expand_dims(idx, -1), axis=-1).squeeze(-1)
    return torch.from_numpy(np.ascontiguousarray(vals)), torch.from_numpy(idx.astype(np.int64))


def _estimators_per_worker(n_estimators: int, world: int) -> List[int]:
    """Tree counts per rank (reference tree.py:330-341): floor division with
    the remainder spread over the first ranks."""
    base = n_estimators // world
    out = [base] * world
    for i in range(n_estimators % world):
        out[i] += 1
    return out


def _resolve_max_features(strategy: str, d: int, task: str) -> int:
    s = str(strategy).lower()
    if s == "auto":
        s = "sqrt" if task == "classification" else "onethird"
    if s == "all":
        return d
    if s == "sqrt":
        return max(1, int(math.sqrt(d)))
    if s == "log2":
        return max(1, int(math.log2(max(2, d))))
    if s == "onethird":
        return max(1, int(d / 3.0))
    try:
        v = float(s)
        if v.is_integer() and v >= 1:
            return min(d, int(v))
        if 0 < v <= 1:
            return max(1, int(v * d))
    except ValueError:
        pass
    raise ValueError(f"unsupported featureSubsetStrategy {strategy!r}")


# ---------------------------------------------------------------------------
# Binning
# ---------------------------------------------------------------------------


def _compute_bin_edges(
    X: torch.Tensor, n_bins: int, comm, sample_cap: int = 65536, seed: int = 0
) -> torch.Tensor:
    """Approximate per-feature quantile edges from a global row sample
    (reference cuML RF quantile binning). Returns [d, n_bins-1] f32 edges."""
    n, d = X.shape
    rng = np.random.default_rng(seed)
    cap_local = max(1, sample_cap // max(1, comm.world_size))
    if n > cap_local:
        idx = torch.from_numpy(rng.choice(n, cap_local, replace=False)).to(X.device)
        sample = X[idx]
    else:
        sample = X
    if comm.world_size > 1:
        parts = comm.allgather_obj(as_numpy(sample))
        full = np.concatenate([p for p in parts if len(p)], axis=0)
        # quantiles by an on-device column sort (np.quantile on a 65k x 3000
        # sample costs seconds on CPU; one torch.sort is milliseconds on GPU)
        ft = torch.from_numpy(full).to(X.device)
    else:
        # single rank: the sample is already device-resident — the pickled
        # allgather round-trip cost ~0.25 s on a 65k x 3000 sample
        ft = sample
    m = ft.shape[0]
    srt, _ = torch.sort(ft, dim=0)
    qpos = (torch.linspace(0, 1, n_bins + 1, device=X.device)[1:-1] * (m - 1)).round().long()
    edges = srt[qpos].T.contiguous().to(torch.float32)  # [d, nb-1]
    edges = torch.cummax(edges, dim=1).values  # monotone per feature
    return edges


def _bin_data(X: torch.Tensor, edges: torch.Tensor) -> torch.Tensor:
    """bin(x) = #edges <= x, i.e. searchsorted right; uint8 [n,d].
    Inference rule derived from this: bin(x) <= b  <=>  x < edges[b].
    Feature-chunked: searchsorted's kernel caps at 2^31 elements and the
    transposed copy stays small."""
    n, d = X.shape
    out = torch.empty((n, d), dtype=torch.uint8, device=X.device)
    fc = max(1, min(d, (1 << 30) // max(1, n)))
    for f0 in range(0, d, fc):
        f1 = min(d, f0 + fc)
        b = torch.searchsorted(
            edges[f0:f1], X[:, f0:f1].T.contiguous(), right=True
        )  # [fc, n]
        out[:, f0:f1] = b.T.to(torch.uint8)
    return out


# ---------------------------------------------------------------------------
# Tree growing
# ---------------------------------------------------------------------------


class _Tree:
    """Flat-array node ARENA with BULK growth (one numpy append per node
    batch — the per-node Python loop was the depth-13 bottleneck). Round 2:
    one arena holds ALL of a worker's trees (cuML-style batched node
    processing, reference tree.py:384-389); `tree_of` records which tree
    owns each node and `split_by_tree` extracts the per-tree arrays."""

    __slots__ = ("feature", "threshold", "left", "right", "is_leaf", "gain", "value", "tree_of", "n_nodes")

    def __init__(self, value_width: int):
        self.feature = [np.empty(0, np.int32)]
        self.threshold = [np.empty(0, np.float32)]
        self.left = [np.empty(0, np.int32)]
        self.right = [np.empty(0, np.int32)]
        self.is_leaf = [np.empty(0, bool)]
        self.gain = [np.empty(0, np.float32)]
        self.value = [np.empty((0, value_width), np.float32)]
        self.tree_of = [np.empty(0, np.int32)]
        self.n_nodes = 0

    def add_leaves(self, values: np.ndarray, tree_ids: Optional[np.ndarray] = None) -> int:
        """Append len(values) leaf nodes; returns the first new node id."""
        m = values.shape[0]
        first = self.n_nodes
        self.n_nodes += m
        self.feature.append(np.full(m, -1, np.int32))
        self.threshold.append(np.zeros(m, np.float32))
        self.left.append(np.full(m, -1, np.int32))
        self.right.append(np.full(m, -1, np.int32))
        self.is_leaf.append(np.ones(m, bool))
        self.gain.append(np.zeros(m, np.float32))
        self.value.append(values.astype(np.float32))
        self.tree_of.append(
            tree_ids.astype(np.int32) if tree_ids is not None else np.zeros(m, np.int32)
        )
        return first

    def _consolidate(self) -> None:
        if len(self.feature) > 1:
            self.feature = [np.concatenate(self.feature)]
            self.threshold = [np.concatenate(self.threshold)]
            self.left = [np.concatenate(self.left)]
            self.right = [np.concatenate(self.right)]
            self.is_leaf = [np.concatenate(self.is_leaf)]
            self.gain = [np.concatenate(self.gain)]
            self.value = [np.concatenate(self.value, axis=0)]
            self.tree_of = [np.concatenate(self.tree_of)]

    def set_splits(
        self,
        node_ids: np.ndarray,
        feature: np.ndarray,
        threshold: np.ndarray,
        left: np.ndarray,
        right: np.ndarray,
        gain: Optional[np.ndarray] = None,
    ) -> None:
        self._consolidate()
        self.feature[0][node_ids] = feature.astype(np.int32)
        self.threshold[0][node_ids] = threshold.astype(np.float32)
        self.left[0][node_ids] = left.astype(np.int32)
        self.right[0][node_ids] = right.astype(np.int32)
        self.is_leaf[0][node_ids] = False
        if gain is not None:
            self.gain[0][node_ids] = gain.astype(np.float32)

    def to_arrays(self) -> Dict[str, np.ndarray]:
        self._consolidate()
        return {
            "feature": self.feature[0],
            "threshold": self.threshold[0],
            "left": self.left[0],
            "right": self.right[0],
            "is_leaf": self.is_leaf[0],
            "gain": self.gain[0],
            "value": self.value[0],
        }

    def split_by_tree(self, n_trees: int) -> List[Dict[str, np.ndarray]]:
        """Extract each tree's arrays from the arena with node ids remapped
        to 0..m-1 (ascending arena order keeps every tree's root at 0:
        roots are created before any child)."""
        self._consolidate()
        tree_of = self.tree_of[0]
        out: List[Dict[str, np.ndarray]] = []
        remap = np.full(self.n_nodes + 1, -1, np.int32)
        for t in range(n_trees):
            ids = np.nonzero(tree_of == t)[0]
            remap[:] = -1
            remap[ids] = np.arange(len(ids), dtype=np.int32)
            left = self.left[0][ids]
            right = self.right[0][ids]
            out.append(
                {
                    "feature": self.feature[0][ids],
                    "threshold": self.threshold[0][ids],
                    "left": np.where(left >= 0, remap[left], -1).astype(np.int32),
                    "right": np.where(right >= 0, remap[right], -1).astype(np.int32),
                    "is_leaf": self.is_leaf[0][ids],
                    "gain": self.gain[0][ids],
                    "value": self.value[0][ids],
                }
            )
        return out


def _grow_forest(
    Xb: torch.Tensor,  # [n,d] uint8 binned (shared across trees)
    y: torch.Tensor,  # [n] int64 class idx (classification) or f32 target
    edges: torch.Tensor,  # [d, nb-1]
    task: str,
    n_classes: int,
    n_bins: int,
    max_depth: int,
    min_leaf: int,
    min_gain: float,
    max_features: int,
    gen: torch.Generator,
    n_trees: int = 1,
    sample: Optional[torch.Tensor] = None,  # int32 [n_trees*per] bootstrap map
    node_batch: int = 16384,
    feat_chunk: int = 256,
) -> List[Dict[str, np.ndarray]]:
    """Grow ALL of this worker's trees at once in one node arena over a
    VIRTUAL row space [n_trees × per] (cuML-style batched node processing —
    reference tree.py:384-389). Virtual row v belongs to tree v // per and
    maps to physical row sample[v] (bootstrap) or v % n. Per-tree loops cost
    trees × depth × batches host round-trips; this costs depth × batches."""
    n, d = Xb.shape
    dev = Xb.device
    value_width = n_classes if task == "classification" else 2  # (mean, count)
    tree = _Tree(value_width)

    per = (sample.numel() // n_trees) if sample is not None else n
    vn = n_trees * per
    t_of_v = torch.arange(vn, device=dev) // max(1, per)
    phys_all = sample.to(torch.int64) if sample is not None else None

    if task == "classification":
        y_idx = y.to(torch.int64)
        if phys_all is not None:
            yp = y_idx[phys_all]
            root_counts = torch.bincount(
                t_of_v * n_classes + yp, minlength=n_trees * n_classes
            ).view(n_trees, n_classes).to(torch.float32)
            root_vals = as_numpy(root_counts)
        else:
            counts = torch.bincount(y_idx, minlength=n_classes).to(torch.float32)
            root_vals = np.tile(as_numpy(counts), (n_trees, 1))
        yf = None
    else:
        yf = y.to(torch.float32)
        if phys_all is not None:
            sums = torch.zeros(n_trees, dtype=torch.float64, device=dev)
            sums.scatter_add_(0, t_of_v, yf[phys_all].to(torch.float64))
            means = as_numpy((sums / max(1, per)).to(torch.float32))
        else:
            means = np.full(n_trees, float(yf.mean().item()) if n else 0.0, np.float32)
        root_vals = np.stack(
            [means, np.full(n_trees, float(per), np.float32)], axis=1
        )
        y_idx = None

    tree.add_leaves(root_vals, tree_ids=np.arange(n_trees, dtype=np.int32))
    edges_np = as_numpy(edges)
    # virtual row v starts at its tree's root (arena ids 0..n_trees-1)
    node_of_row = t_of_v.clone()
    sample_t = (
        sample.contiguous()
        if sample is not None
        else torch.empty(0, dtype=torch.int32, device=dev)
    )

    # HIP histogram path (LDS-privatized, rows segment-sorted by node)
    import os as _os

    from ..ops.dispatch import has_hip_ops, use_hip

    hip_hist = Xb.is_cuda and use_hip(Xb) and has_hip_ops()
    _rf_debug = _os.environ.get("SRML_RF_DEBUG") == "1"
    _rf_t = _os.environ.get("SRML_RF_DEBUG") == "2"
    _acc: Dict[str, float] = {}

    def _tick(label, t0):
        if not _rf_t:
            return 0.0
        import time as _tm

        if Xb.is_cuda:
            torch.cuda.synchronize()
        t = _tm.perf_counter()
        _acc[label] = _acc.get(label, 0.0) + (t - t0)
        return t

    def _now():
        if not _rf_t:
            return 0.0
        import time as _tm

        if Xb.is_cuda:
            torch.cuda.synchronize()
        return _tm.perf_counter()
    if _rf_debug:
        print(f"[rf-debug] hip_hist={hip_hist} n={n} d={d} device={dev}", flush=True)
    if hip_hist:
        from ..ops.dispatch import hip_ops

        ext = hip_ops()
        y32 = y_idx.to(torch.int32) if task == "classification" else yf
        # max|y| for the packed regression histogram cells, ONE sync per fit
        y_max = (
            float(yf.abs().max().item()) if task == "regression" and n else 0.0
        )
        if y_max <= 0.0 and task == "regression":
            y_max = 1.0
        C_ch = n_classes if task == "classification" else 2
        # Measured dispatch (A/B in profiles/README.md): the row-lane
        # kernel wins for classification (sqrt-sampled features, sorted
        # bootstrap = dense segments), the feature-wide kernel wins for
        # regression (1/3-sampled features); SRML_RF_HIST=old|fw overrides
        _mode = _os.environ.get("SRML_RF_HIST")
        _hist_old = (_mode == "old") or (_mode != "fw" and task == "classification")
        if _hist_old:
            fc_kernel = max(1, min(512, (30 * 1024 // 4) // max(1, n_bins * C_ch)))
        else:
            fc_kernel = max(1, min(32, (150 * 1024 // 4) // max(1, n_bins * C_ch)))
        # column-major binned matrix for the histogram/reroute kernels: a
        # node segment's row gathers stay inside dense per-feature cache
        # lines (row-major fetched a ~47-line row to read ~54 sampled bytes)
        Xcm = Xb.T.contiguous()
    frontier = list(range(n_trees))  # arena node ids still splittable

    for depth in range(max_depth):
        if not frontier:
            break
        new_frontier: List[int] = []
        for bstart in range(0, len(frontier), node_batch):
            batch = frontier[bstart : bstart + node_batch]
            B = len(batch)
            batch_t = torch.tensor(batch, dtype=torch.int64, device=dev)
            # map node id -> local 0..B-1 (rows not in batch -> -1)
            lut = torch.full((tree.n_nodes,), -1, dtype=torch.int64, device=dev)
            lut[batch_t] = torch.arange(B, dtype=torch.int64, device=dev)
            if hip_hist:
                _t0 = _now()
                # counting-sort partition kernel: one pass for counts, one
                # scatter — replaces sort+nonzero+gathers (rf_partition)
                perm, seg_off = ext.rf_partition(node_of_row, lut, B)
                _tick("partition", _t0)
                # no host sync here: an all-empty batch flows through as
                # no-op kernels and an empty split set
                rows = loc = yb = None
            else:
                local = lut[node_of_row]
                in_batch = local >= 0
                rows = torch.nonzero(in_batch).flatten()
                if rows.numel() == 0:
                    continue
                loc = local[rows]
                # virtual -> physical rows for feature/label gathers
                prows = phys_all[rows] if phys_all is not None else rows % n
                if task == "classification":
                    yb = y_idx[prows]
                else:
                    yb = yf[prows]

            best_gain = torch.full((B,), -1.0, dtype=torch.float32, device=dev)
            best_feat = torch.full((B,), -1, dtype=torch.int64, device=dev)
            best_bin = torch.full((B,), -1, dtype=torch.int64, device=dev)
            best_lcnt = torch.zeros((B,), dtype=torch.float32, device=dev)
            if task == "classification":
                best_lval = torch.zeros((B, n_classes), dtype=torch.float32, device=dev)
                best_rval = torch.zeros((B, n_classes), dtype=torch.float32, device=dev)
            else:
                best_lval = torch.zeros((B, 2), dtype=torch.float32, device=dev)
                best_rval = torch.zeros((B, 2), dtype=torch.float32, device=dev)

            # per-node sampled feature subset: histograms are built ONLY for
            # sampled features (a gather per node), not all d then masked —
            # at sqrt(d) sampling this is the d/sqrt(d)-fold work saver the
            # reference gets from cuML's per-node sampling.
            _t0 = _now()
            if max_features < d:
                scores = torch.rand((B, d), generator=gen, device=dev)
                # uniform subset without replacement: topk of random scores
                # for small mf; torch.topk degrades badly past k~256 on ROCm
                # (389 ms for one [10,3000] k=1000 call) — radix argsort wins
                if max_features <= 256:
                    feat_sel = scores.topk(max_features, dim=1).indices
                else:
                    feat_sel = scores.argsort(dim=1)[:, :max_features]
                if hip_hist and not _hist_old:
                    # feature-wide kernel wants SORTED per-node subsets: a
                    # wave-load then spans consecutive row bytes (avg gap
                    # d/mf) instead of the whole row
                    feat_sel = feat_sel.sort(dim=1).values
                _tick("feat_sel", _t0)
                mf = max_features
                Xb_rows = None  # sampled path reads only selected bytes
            else:
                feat_sel = None
                mf = d
                Xb_rows = None
            if hip_hist:
                fsel32 = (
                    feat_sel.to(torch.int32).contiguous()
                    if feat_sel is not None
                    else torch.empty((0, 0), dtype=torch.int32, device=dev)
                )
                fc_step = fc_kernel
            else:
                if feat_sel is None:
                    Xb_rows = Xb[prows]
                fc_step = feat_chunk
            _t0 = _now()
            for f0 in range(0, mf, fc_step):
                f1 = min(mf, f0 + fc_step)
                F = f1 - f0
                if hip_hist:
                    _hist_fn = ext.rf_histogram if _hist_old else ext.rf_histogram_fw
                    _hist_x = Xcm if _hist_old else Xb
                    H = _hist_fn(
                        _hist_x, perm, seg_off, fsel32, y32, f0, F, n_bins,
                        n_classes if task == "classification" else 0,
                        sample_t, y_max if task == "regression" else 0.0,
                    )
                    if H.shape[3] <= 16:
                        # fused gain scan + per-node best (kernel)
                        g, fidx, sbin_b, lval_b, rval_b = ext.rf_best_split(
                            H, min_leaf, task == "classification"
                        )
                        fidx = fidx.to(torch.int64)
                        upd = g > best_gain
                        best_gain = torch.where(upd, g, best_gain)
                        ar = torch.arange(B, device=dev)
                        if feat_sel is not None:
                            chosen = feat_sel[ar, (fidx + f0).clamp(min=0)]
                        else:
                            chosen = fidx + f0
                        best_feat = torch.where(upd, chosen, best_feat)
                        best_bin = torch.where(upd, sbin_b.to(torch.int64), best_bin)
                        if task == "regression":
                            # kernel emits raw (count, sum); the tree stores
                            # (mean, count) leaf values
                            lc = torch.clamp(lval_b[:, 0], min=1e-12)
                            rc = torch.clamp(rval_b[:, 0], min=1e-12)
                            lval_b = torch.stack([lval_b[:, 1] / lc, lval_b[:, 0]], dim=1)
                            rval_b = torch.stack([rval_b[:, 1] / rc, rval_b[:, 0]], dim=1)
                        best_lval = torch.where(upd[:, None], lval_b, best_lval)
                        best_rval = torch.where(upd[:, None], rval_b, best_rval)
                        continue
                    if task == "classification":
                        gain, sbin, lval, rval, lcnt = _best_split_class(H, min_leaf)
                    else:
                        gain, sbin, lval, rval, lcnt = _best_split_reg(H, min_leaf)
                elif task == "classification":
                    if feat_sel is not None:
                        sel = feat_sel[loc][:, f0:f1]
                        bins = Xb[prows[:, None], sel].to(torch.int64)
                    else:
                        bins = Xb_rows[:, f0:f1].to(torch.int64)
                    base = (loc[:, None] * F + torch.arange(F, device=dev)[None, :]) * n_bins + bins
                    hist = torch.zeros(B * F * n_bins * n_classes, dtype=torch.float32, device=dev)
                    idx = base * n_classes + yb[:, None]
                    hist.index_add_(0, idx.flatten(), torch.ones(idx.numel(), device=dev))
                    H = hist.view(B, F, n_bins, n_classes)
                    gain, sbin, lval, rval, lcnt = _best_split_class(H, min_leaf)
                else:
                    if feat_sel is not None:
                        sel = feat_sel[loc][:, f0:f1]
                        bins = Xb[prows[:, None], sel].to(torch.int64)
                    else:
                        bins = Xb_rows[:, f0:f1].to(torch.int64)
                    base = (loc[:, None] * F + torch.arange(F, device=dev)[None, :]) * n_bins + bins
                    # (count, sum) only: the sum-of-squares terms cancel in
                    # the variance-reduction gain (ls²/lc + rs²/rc - ts²/tc)
                    hist2 = torch.zeros(B * F * n_bins, 2, dtype=torch.float32, device=dev)
                    flat = base.flatten()
                    src = torch.empty(bins.shape[0], 2, dtype=torch.float32, device=dev)
                    src[:, 0] = 1.0
                    src[:, 1] = yb
                    hist2.index_add_(0, flat, src.repeat_interleave(F, dim=0) if F > 1 else src)
                    H = hist2.view(B, F, n_bins, 2)
                    gain, sbin, lval, rval, lcnt = _best_split_reg(H, min_leaf)
                # best feature within chunk
                g, fidx = _max_lastdim(gain)
                upd = g > best_gain
                best_gain = torch.where(upd, g, best_gain)
                ar = torch.arange(B, device=dev)
                if feat_sel is not None:
                    chosen = feat_sel[ar, fidx + f0]
                else:
                    chosen = fidx + f0
                best_feat = torch.where(upd, chosen, best_feat)
                best_bin = torch.where(upd, sbin[ar, fidx], best_bin)
                best_lcnt = torch.where(upd, lcnt[ar, fidx], best_lcnt)
                best_lval = torch.where(upd[:, None], lval[ar, fidx], best_lval)
                best_rval = torch.where(upd[:, None], rval[ar, fidx], best_rval)

            _tick("hist_scan", _t0)
            _t0 = _now()
            # materialize splits: fully vectorized (one bulk leaf append +
            # one set_splits per batch; the per-node loop was the depth-13
            # wall-clock bottleneck)
            # ONE packed D2H instead of five device syncs (values fit f32
            # exactly: features < 2^24, bins < 256)
            pack = torch.cat(
                [
                    best_gain[:, None],
                    best_feat.to(torch.float32)[:, None],
                    best_bin.to(torch.float32)[:, None],
                    best_lval,
                    best_rval,
                ],
                dim=1,
            )
            packed = as_numpy(pack)
            w = best_lval.shape[1]
            bg = packed[:, 0]
            bf = packed[:, 1].astype(np.int64)
            bb = packed[:, 2].astype(np.int64)
            lv = packed[:, 3 : 3 + w]
            rv = packed[:, 3 + w :]
            batch_np = np.asarray(batch, dtype=np.int64)
            valid = (bg > min_gain) & (bf >= 0)
            idxs = np.nonzero(valid)[0]
            if len(idxs) == 0:
                continue
            ns = len(idxs)
            children = np.empty((2 * ns, lv.shape[1]), dtype=np.float32)
            children[0::2] = lv[idxs]
            children[1::2] = rv[idxs]
            # children inherit their parent's tree
            tree._consolidate()
            ptree = tree.tree_of[0][batch_np[idxs]]
            first = tree.add_leaves(children, tree_ids=np.repeat(ptree, 2))
            l_ids = first + 2 * np.arange(ns)
            r_ids = l_ids + 1
            fsel = bf[idxs].astype(np.int64)
            bsel = bb[idxs].astype(np.int64)
            tree.set_splits(
                batch_np[idxs], fsel, edges_np[fsel, bsel], l_ids, r_ids,
                gain=bg[idxs],
            )
            if depth + 1 < max_depth:
                if task == "regression":
                    lcount = lv[idxs][:, -1]
                    rcount = rv[idxs][:, -1]
                else:
                    lcount = lv[idxs].sum(axis=1)
                    rcount = rv[idxs].sum(axis=1)
                new_frontier.extend(l_ids[lcount >= 2 * min_leaf].tolist())
                new_frontier.extend(r_ids[rcount >= 2 * min_leaf].tolist())
            _tick("materialize", _t0)
            _t0 = _now()
            # reroute rows of split nodes (touch only the split feature byte)
            nid_t = torch.from_numpy(batch_np[idxs]).to(dev)
            f_t = torch.from_numpy(fsel).to(dev)
            b_t = torch.from_numpy(bsel).to(dev)
            l_t = torch.from_numpy(l_ids).to(dev)
            r_t = torch.from_numpy(r_ids).to(dev)
            lut2 = torch.full((tree.n_nodes,), -1, dtype=torch.int64, device=dev)
            lut2[nid_t] = torch.arange(ns, dtype=torch.int64, device=dev)
            if hip_hist:
                # single-pass kernel: gather split byte + child write fused
                ext.rf_reroute(
                    node_of_row, lut2,
                    f_t.to(torch.int32), b_t.to(torch.int32), l_t, r_t, Xcm,
                    sample_t,
                )
            else:
                sl = lut2[node_of_row]
                mrows = torch.nonzero(sl >= 0).flatten()
                srel = sl[mrows]
                pm = phys_all[mrows] if phys_all is not None else mrows % n
                go_left = Xb[pm, f_t[srel]].to(torch.int64) <= b_t[srel]
                node_of_row[mrows] = torch.where(go_left, l_t[srel], r_t[srel])
            _tick("reroute", _t0)
        if _rf_t:
            pass
        if _rf_debug:
            print(
                f"[rf-debug] depth={depth} frontier={len(frontier)} -> "
                f"{len(new_frontier)} nodes={tree.n_nodes}",
                flush=True,
            )
        frontier = new_frontier
    if _rf_t:
        print(f"[rf-times] {sorted(_acc.items(), key=lambda kv: -kv[1])}", flush=True)

    return tree.split_by_tree(n_trees)


def _best_split_class(H: torch.Tensor, min_leaf: int):
    """H: [B,F,nb,C] class counts. Returns per (B,F): best gain (impurity
    decrease weighted by node fraction, Spark/CART gini), split bin, left/right
    class-count values, left count."""
    csum = H.cumsum(dim=2)  # left counts for split at bin b (bins <= b)
    total = csum[:, :, -1:, :]  # [B,F,1,C]
    left = csum[:, :, :-1, :]  # split bins 0..nb-2
    right = total - left
    lc = left.sum(dim=3)
    rc = right.sum(dim=3)
    nt = total.sum(dim=3)  # [B,F,1]
    gini = lambda cnt, tot: 1.0 - ((cnt / torch.clamp(tot.unsqueeze(-1), min=1e-12)) ** 2).sum(
        dim=-1
    )
    g_parent = gini(total, nt)  # [B,F,1]
    g_left = gini(left, lc)
    g_right = gini(right, rc)
    ntc = torch.clamp(nt, min=1e-12)
    gain = g_parent - (lc / ntc) * g_left - (rc / ntc) * g_right  # [B,F,nb-1]
    valid = (lc >= min_leaf) & (rc >= min_leaf)
    gain = torch.where(valid, gain, torch.full_like(gain, -1.0))
    best_gain, best_bin = _max_lastdim(gain)  # [B,F]
    B, F = best_gain.shape
    ar_b = torch.arange(B, device=H.device)[:, None].expand(B, F)
    ar_f = torch.arange(F, device=H.device)[None, :].expand(B, F)
    lval = left[ar_b, ar_f, best_bin]  # [B,F,C]
    rval = right[ar_b, ar_f, best_bin]
    lcnt = lc[ar_b, ar_f, best_bin]
    return best_gain, best_bin, lval, rval, lcnt


def _best_split_reg(H: torch.Tensor, min_leaf: int):
    """H: [B,F,nb,2] = (count,sum). Variance-reduction gain — the
    sum-of-squares terms cancel across parent/children, leaving
    (ls²/lc + rs²/rc - ts²/tc)/tc, so the histogram needs no y² channel."""
    cs = H.cumsum(dim=2)
    total = cs[:, :, -1:, :]
    left = cs[:, :, :-1, :]
    right = total - left
    lc, ls = left[..., 0], left[..., 1]
    rc, rs = right[..., 0], right[..., 1]
    tc, ts = total[..., 0], total[..., 1]
    lcc = torch.clamp(lc, min=1e-12)
    rcc = torch.clamp(rc, min=1e-12)
    tcc = torch.clamp(tc, min=1e-12)
    gain = (ls * ls / lcc + rs * rs / rcc - ts * ts / tcc) / tcc
    valid = (lc >= min_leaf) & (rc >= min_leaf)
    gain = torch.where(valid, gain, torch.full_like(gain, -1.0))
    best_gain, best_bin = _max_lastdim(gain)
    B, F = best_gain.shape
    ar_b = torch.arange(B, device=H.device)[:, None].expand(B, F)
    ar_f = torch.arange(F, device=H.device)[None, :].expand(B, F)
    lcb = lc[ar_b, ar_f, best_bin]
    lsb = ls[ar_b, ar_f, best_bin]
    rcb = rc[ar_b, ar_f, best_bin]
    rsb = rs[ar_b, ar_f, best_bin]
    lval = torch.stack([lsb / torch.clamp(lcb, min=1e-12), lcb], dim=-1)
    rval = torch.stack([rsb / torch.clamp(rcb, min=1e-12), rcb], dim=-1)
    return best_gain, best_bin, lval, rval, lcb


# ---------------------------------------------------------------------------
# Estimator / Model
# ---------------------------------------------------------------------------


class _RandomForestEstimator(_RandomForestParams, Estimator):
    _task = "regression"

    def setNumTrees(self, value: int):
        return self._set_params(numTrees=value)

    def setMaxDepth(self, value: int):
        return self._set_params(maxDepth=value)

    def setMaxBins(self, value: int):
        return self._set_params(maxBins=value)

    def setSeed(self, value: int):
        return self._set_params(seed=value)

    def setImpurity(self, value: str):
        return self._set_params(impurity=value)

    def setFeaturesCol(self, value):
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setLabelCol(self, value: str):
        return self._set_params(labelCol=value)

    def setPredictionCol(self, value: str):
        return self._set_params(predictionCol=value)

    def setFeatureSubsetStrategy(self, value: str):
        return self._set_params(featureSubsetStrategy=value)

    def _fit_array(
        self, X: Any, y: Optional[Any], ctx: _FitContext, params: Dict[str, Any]
    ) -> Dict[str, Any]:
        comm, pdesc = ctx.comm, ctx.pdesc
        n_estimators = int(params["n_estimators"])
        max_depth = int(params["max_depth"])
        n_bins = min(256, int(params["n_bins"]))
        min_leaf = int(params["min_samples_leaf"])
        min_gain = float(params["min_impurity_decrease"])
        seed = int(params["random_state"])
        bootstrap = bool(params["bootstrap"])
        max_samples = float(params["max_samples"])

        import os as _os
        import time as _time

        _dbg = _os.environ.get("SRML_RF_DEBUG") == "1"

        def _phase(label: str, t0: float) -> float:
            if _dbg:
                if ctx.device.type == "cuda":
                    torch.cuda.synchronize(ctx.device)
                t = _time.perf_counter()
                print(f"[rf-phase] {label}: {t - t0:.3f}s", flush=True)
                return t
            return t0

        _t = _time.perf_counter()
        Xt = ctx.device_tensor(np.asarray(X, dtype=np.float32))
        n, d = Xt.shape
        _t = _phase("ingest_h2d", _t)

        if self._task == "classification":
            y_local = np.asarray(y)
            bad = (y_local < 0) | (y_local != np.floor(y_local))
            if bad.any():
                raise ValueError("RandomForestClassifier labels must be non-negative integers")
            local_max = float(y_local.max()) if len(y_local) else -1.0
            n_classes = int(comm.allreduce_scalar(local_max, "max")) + 1
            # reference requires labels to cover [0, numClasses) (tree.py:415-421)
            yt = to_device_tensor(y_local.astype(np.int64), ctx.device)
        else:
            n_classes = 0
            yt = to_device_tensor(np.asarray(y, dtype=np.float32), ctx.device)

        edges = _compute_bin_edges(Xt, n_bins, comm, seed=seed)
        _t = _phase("bin_edges", _t)
        Xb = _bin_data(Xt, edges)
        _t = _phase("bin_data", _t)
        max_features = _resolve_max_features(str(params["max_features"]), d, self._task)

        counts = _estimators_per_worker(n_estimators, comm.world_size)
        my_trees = counts[comm.rank]
        tree_id0 = sum(counts[: comm.rank])

        trees: List[Dict[str, np.ndarray]] = []
        if my_trees > 0 and n > 0:
            # all of this worker's trees grow together in ONE arena over a
            # virtual row space (bootstrap = int32 sample map, no gathered
            # [n,d] copies per tree); seed offset by tree_id0 so ranks draw
            # disjoint streams. The virtual row space costs ~13 bytes/row
            # (node map + sample), so cap it at ~5e8 rows and grow LARGE
            # forests in sequential tree groups (500 trees × 20M rows would
            # otherwise need >100 GB of arena state).
            group_cap = max(1, int(5e8) // max(1, n))
            t0g = 0
            while t0g < my_trees:
                g_trees = min(group_cap, my_trees - t0g)
                gen = torch.Generator(device=ctx.device)
                gen.manual_seed(
                    (seed + 1315423911 * (tree_id0 + t0g + 1)) & 0x7FFFFFFFFFFF
                )
                if bootstrap:
                    n_draw = max(1, int(round(max_samples * n)))
                    sample = torch.randint(
                        0, n, (g_trees * n_draw,), generator=gen,
                        device=ctx.device, dtype=torch.int32,
                    )
                    # sort each tree's draw: a bootstrap multiset is
                    # unordered, and ascending physical rows turn the
                    # histogram's random per-row byte gathers into
                    # near-sequential column reads. ONE flat radix sort over
                    # tree-offset keys (the segmented [T, n] sort.values
                    # cost 390 ms at 10x1M)
                    t_of = torch.arange(
                        g_trees, device=ctx.device
                    ).repeat_interleave(n_draw)
                    keys = t_of * n + sample.to(torch.int64)
                    keys, _ = torch.sort(keys)
                    sample = (keys % n).to(torch.int32)
                else:
                    sample = None
                trees.extend(
                    _grow_forest(
                        Xb,
                        yt,
                        edges,
                        self._task,
                        n_classes,
                        n_bins,
                        max_depth,
                        min_leaf,
                        min_gain,
                        max_features,
                        gen,
                        n_trees=g_trees,
                        sample=sample,
                        node_batch=int(params.get("max_batch_size", 4096)),
                    )
                )
                t0g += g_trees
        elif my_trees > 0:
            # empty local shard: contribute degenerate single-leaf trees so
            # the merged forest still has numTrees members
            width = n_classes if self._task == "classification" else 2
            trees = [
                {
                    "feature": np.full(1, -1, np.int32),
                    "threshold": np.zeros(1, np.float32),
                    "left": np.full(1, -1, np.int32),
                    "right": np.full(1, -1, np.int32),
                    "is_leaf": np.ones(1, bool),
                    "gain": np.zeros(1, np.float32),
                    "value": np.zeros((1, width), np.float32),
                }
                for _ in range(my_trees)
            ]
        _t = _phase("grow_trees_total", _t)

        _t = _phase("grow_trees", _t)
        # merge sub-forests (reference allGathers treelite bytes, tree.py:424-447)
        blobs = comm.allgather_obj(pickle.dumps(trees, protocol=4))
        forest: List[Dict[str, np.ndarray]] = []
        for b in blobs:
            forest.extend(pickle.loads(b))

        return {
            "forest_": np.frombuffer(pickle.dumps(forest, protocol=4), dtype=np.uint8).copy(),
            "n_classes_": n_classes,
            "n_features_": d,
            "n_trees_": len(forest),
        }


class _RandomForestModel(_RandomForestParams, Model):
    _task = "regression"

    def __init__(
        self,
        forest_: np.ndarray,
        n_classes_: int = 0,
        n_features_: int = 0,
        n_trees_: int = 0,
        **kwargs: Any,
    ) -> None:
        super().__init__(
            forest_=np.asarray(forest_, dtype=np.uint8),
            n_classes_=int(n_classes_),
            n_features_=int(n_features_),
            n_trees_=int(n_trees_),
        )
        self._trees: Optional[List[Dict[str, np.ndarray]]] = None

    @property
    def trees(self) -> List[Dict[str, np.ndarray]]:
        if self._trees is None:
            self._trees = pickle.loads(self._model_attributes["forest_"].tobytes())
        return self._trees

    @property
    def numTrees(self) -> int:
        return self._model_attributes["n_trees_"]

    @property
    def numFeatures(self) -> int:
        return self._model_attributes["n_features_"]

    @property
    def numClasses(self) -> int:
        return self._model_attributes["n_classes_"]

    @property
    def treeWeights(self) -> List[float]:
        return [1.0] * self.numTrees

    @property
    def totalNumNodes(self) -> int:
        return int(sum(t["feature"].shape[0] for t in self.trees))

    @property
    def featureImportances(self) -> np.ndarray:
        """Impurity-decrease feature importances (Spark semantics: per-tree
        gain x node-count sums, normalized per tree then averaged)."""
        total = np.zeros(self.numFeatures)
        for t in self.trees:
            imp = np.zeros(self.numFeatures)
            internal = ~t["is_leaf"]
            if self._task == "classification":
                counts = t["value"].sum(axis=1)
            else:
                counts = t["value"][:, 1]
            f = t["feature"][internal]
            w = t.get("gain", np.zeros(len(t["feature"])))[internal] * counts[internal]
            np.add.at(imp, f, w)
            tot = imp.sum()
            if tot > 0:
                total += imp / tot
        tot = total.sum()
        return total / tot if tot > 0 else total

    def toDebugString(self) -> str:
        """Human-readable forest dump (Spark toDebugString parity)."""
        lines = [f"{type(self).__name__} with {self.numTrees} trees"]
        for ti, t in enumerate(self.trees):
            lines.append(f"  Tree {ti} ({t['feature'].shape[0]} nodes):")

            def rec(node: int, depth: int) -> None:
                pad = "    " * (depth + 1)
                if t["is_leaf"][node]:
                    lines.append(f"{pad}Predict: {t['value'][node].tolist()}")
                else:
                    lines.append(
                        f"{pad}If (feature {t['feature'][node]} < "
                        f"{t['threshold'][node]:.6g})"
                    )
                    rec(int(t["left"][node]), depth + 1)
                    lines.append(f"{pad}Else")
                    rec(int(t["right"][node]), depth + 1)

            if t["feature"].shape[0] <= 2047:  # bound output size
                rec(0, 1)
        return "\n".join(lines)

    def dump_as_json(self) -> str:
        """Forest dump for Spark-tree translation parity (reference
        treelite dump_as_json, tree.py:449-460)."""
        import json

        return json.dumps(
            [
                {k: v.tolist() for k, v in t.items()}
                for t in self.trees
            ]
        )

    def setFeaturesCol(self, value):
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setPredictionCol(self, value: str):
        return self._set_params(predictionCol=value)

    def _device_arena(self, device) -> Dict[str, torch.Tensor]:
        """Flatten all trees into one device node arena for fil_predict
        (cached on the model; rebuilt only on device change)."""
        cached = getattr(self, "_fil_arena", None)
        if cached is not None and cached["feat"].device == device:
            return cached
        feats, thrs, lefts, rights, vals, roots = [], [], [], [], [], []
        off = 0
        for t in self.trees:
            m = t["feature"].shape[0]
            roots.append(off)
            # leaves marked by feature = -1 (the kernel's stop test)
            f = np.where(t["is_leaf"], -1, t["feature"]).astype(np.int32)
            feats.append(f)
            thrs.append(t["threshold"].astype(np.float32))
            lefts.append(t["left"].astype(np.int32))
            rights.append(t["right"].astype(np.int32))
            vals.append(np.atleast_2d(t["value"]).astype(np.float32))
            off += m
        arena = {
            "feat": torch.from_numpy(np.concatenate(feats)).to(device),
            "thr": torch.from_numpy(np.concatenate(thrs)).to(device),
            "left": torch.from_numpy(np.concatenate(lefts)).to(device),
            "right": torch.from_numpy(np.concatenate(rights)).to(device),
            "value": torch.from_numpy(np.concatenate(vals, axis=0)).contiguous().to(device),
            "roots": torch.tensor(roots, dtype=torch.int64, device=device),
        }
        self._fil_arena = arena
        return arena

    def _predict_raw(self, X: Any) -> torch.Tensor:
        """Forest inference. GPU: the fil_predict HIP kernel — one thread
        walks every tree for its row over a flattened node arena (the
        reference's FIL predict, tree.py:709-721). CPU fallback: vectorized
        level-wise torch traversal. Returns [n, C] vote sums
        (classification) or [n] mean."""
        from ..ops.dispatch import has_hip_ops, hip_ops, use_hip
        from ..parallel.context import get_comm

        device = get_comm().device
        Xt = to_device_tensor(np.ascontiguousarray(X, dtype=np.float32), device)
        n = Xt.shape[0]
        vw = self.numClasses if self._task == "classification" else 2
        if (
            Xt.is_cuda
            and use_hip(Xt)
            and has_hip_ops()
            and 0 < vw <= 16
            and n > 0
            and self.numTrees > 0
        ):
            a = self._device_arena(device)
            out = hip_ops().fil_predict(
                Xt.contiguous(), a["feat"], a["thr"], a["left"], a["right"],
                a["value"], a["roots"], self._task == "classification",
            )
            if self._task == "classification":
                return out
            return out[:, 0]
        if self._task == "classification":
            acc = torch.zeros((n, self.numClasses), dtype=torch.float32, device=device)
        else:
            acc = torch.zeros(n, dtype=torch.float32, device=device)
        for t in self.trees:
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
                # strict <: training partitions on bin(x) <= b  <=>  x <
                # edges[b] (=stored threshold), so rows equal to the
                # threshold went RIGHT during fit
                go_left = xv < thr[node]
                nxt = torch.where(go_left, left[node], right[node])
                node = torch.where(at_leaf, node, nxt)
            v = value[node]
            if self._task == "classification":
                acc += v / torch.clamp(v.sum(dim=1, keepdim=True), min=1e-12)
            else:
                acc += v[:, 0]
        return acc

    def predictLeaf(self, value) -> np.ndarray:
        """Leaf ordinal per tree for one feature vector (reference
        tree.py:613-617 delegates to the Spark model; here a host
        traversal — leaves numbered left-to-right per tree)."""
        x = np.asarray(value, dtype=np.float32).ravel()
        out = np.zeros(self.numTrees, dtype=np.float64)
        for ti, t in enumerate(self.trees):
            # leaf ordinals by ascending node id among leaves
            leaf_ord = np.cumsum(t["is_leaf"]) - 1
            node = 0
            while not t["is_leaf"][node]:
                f = int(t["feature"][node])
                node = int(
                    t["left"][node] if x[f] < t["threshold"][node] else t["right"][node]
                )
            out[ti] = float(leaf_ord[node])
        return out

    def predict(self, value) -> float:
        """Single-vector prediction (pyspark RandomForest*Model.predict)."""
        out = self._transform_array(np.asarray(value, dtype=np.float32).reshape(1, -1))
        if isinstance(out, dict):
            return float(out[self.getOrDefault("predictionCol")][0])
        return float(out[0])

    def predictProbability(self, value) -> np.ndarray:
        if self._task != "classification":
            raise AttributeError("predictProbability is classification-only")
        out = self._transform_array(np.asarray(value, dtype=np.float32).reshape(1, -1))
        return np.asarray(out[self.getOrDefault("probabilityCol")][0])

    def predictRaw(self, value) -> np.ndarray:
        if self._task != "classification":
            raise AttributeError("predictRaw is classification-only")
        out = self._transform_array(np.asarray(value, dtype=np.float32).reshape(1, -1))
        return np.asarray(out[self.getOrDefault("rawPredictionCol")][0])

    def _transform_array(self, X: Any):
        acc = self._predict_raw(X)
        if self._task == "classification":
            probs = acc / self.numTrees
            pred = probs.argmax(dim=1)
            out = {
                self.getOrDefault("predictionCol"): as_numpy(pred).astype(np.float64),
            }
            if self.hasParam("probabilityCol"):
                out[self.getOrDefault("probabilityCol")] = as_numpy(probs)
            if self.hasParam("rawPredictionCol"):
                out[self.getOrDefault("rawPredictionCol")] = as_numpy(acc)
            return out
        return as_numpy(acc / self.numTrees).astype(np.float64)
