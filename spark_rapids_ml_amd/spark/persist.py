"""Apache Spark ML on-disk model format, written/read without a JVM.

Layout (what stock Spark's DefaultParamsWriter/Reader and the per-model
readers produce/consume — reference models expose the same directories via
Spark itself; reference core.py:268-355 persists its own side-car format and
relies on `cpu()` + py4j (utils.py:579-809) for Spark-loadable output):

    <path>/metadata/part-00000     one-line JSON: class/timestamp/
                                   sparkVersion/uid/paramMap/defaultParamMap
                                   (+ model-specific top-level extras)
    <path>/data/*.parquet          model payload; ml.linalg Vector/Matrix
                                   columns are UDT structs, and the parquet
                                   FOOTER carries the Spark schema JSON under
                                   "org.apache.spark.sql.parquet.row.metadata"
                                   so Spark's reader restores the UDTs
    <path>/treesMetadata/*.parquet (forests) per-tree metadata JSON + weights

Supported model classes (save + load):
    org.apache.spark.ml.clustering.KMeansModel
    org.apache.spark.ml.feature.PCAModel
    org.apache.spark.ml.regression.LinearRegressionModel
    org.apache.spark.ml.classification.LogisticRegressionModel
    org.apache.spark.ml.classification.RandomForestClassificationModel
    org.apache.spark.ml.regression.RandomForestRegressionModel

Caveats (documented deviations):
- RF regression impurityStats are reconstructed as [count, mean*count,
  mean^2*count] (per-node sum-of-squares is not retained after fit), so the
  reloaded tree's displayed node impurity is 0; predictions are exact.
- RF split thresholds are exported as nextafter(thr, -inf): this engine
  routes left on x < thr while Spark routes left on x <= thr.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

SPARK_VERSION = "3.5.1"

# ---------------------------------------------------------------------------
# Spark schema JSON fragments (org.apache.spark.sql.types DataType.json)
# ---------------------------------------------------------------------------

_VECTOR_SQL_TYPE = {
    "type": "struct",
    "fields": [
        {"name": "type", "type": "byte", "nullable": False, "metadata": {}},
        {"name": "size", "type": "integer", "nullable": True, "metadata": {}},
        {
            "name": "indices",
            "type": {"type": "array", "elementType": "integer", "containsNull": False},
            "nullable": True,
            "metadata": {},
        },
        {
            "name": "values",
            "type": {"type": "array", "elementType": "double", "containsNull": False},
            "nullable": True,
            "metadata": {},
        },
    ],
}

VECTOR_UDT = {
    "type": "udt",
    "class": "org.apache.spark.ml.linalg.VectorUDT",
    "pyClass": "pyspark.ml.linalg.VectorUDT",
    "sqlType": _VECTOR_SQL_TYPE,
}

_MATRIX_SQL_TYPE = {
    "type": "struct",
    "fields": [
        {"name": "type", "type": "byte", "nullable": False, "metadata": {}},
        {"name": "numRows", "type": "integer", "nullable": False, "metadata": {}},
        {"name": "numCols", "type": "integer", "nullable": False, "metadata": {}},
        {
            "name": "colPtrs",
            "type": {"type": "array", "elementType": "integer", "containsNull": False},
            "nullable": True,
            "metadata": {},
        },
        {
            "name": "rowIndices",
            "type": {"type": "array", "elementType": "integer", "containsNull": False},
            "nullable": True,
            "metadata": {},
        },
        {
            "name": "values",
            "type": {"type": "array", "elementType": "double", "containsNull": False},
            "nullable": True,
            "metadata": {},
        },
        {"name": "isTransposed", "type": "boolean", "nullable": False, "metadata": {}},
    ],
}

MATRIX_UDT = {
    "type": "udt",
    "class": "org.apache.spark.ml.linalg.MatrixUDT",
    "pyClass": "pyspark.ml.linalg.MatrixUDT",
    "sqlType": _MATRIX_SQL_TYPE,
}


def _f(name: str, dtype: Any, nullable: bool = True) -> Dict[str, Any]:
    return {"name": name, "type": dtype, "nullable": nullable, "metadata": {}}


def _schema_json(fields: List[Dict[str, Any]]) -> str:
    return json.dumps({"type": "struct", "fields": fields}, separators=(",", ":"))


# ---------------------------------------------------------------------------
# pyarrow builders for UDT columns
# ---------------------------------------------------------------------------


def _pa():
    import pyarrow as pa

    return pa


def _vector_struct_type():
    pa = _pa()
    return pa.struct(
        [
            pa.field("type", pa.int8(), nullable=False),
            pa.field("size", pa.int32()),
            pa.field("indices", pa.list_(pa.int32())),
            pa.field("values", pa.list_(pa.float64())),
        ]
    )


def _matrix_struct_type():
    pa = _pa()
    return pa.struct(
        [
            pa.field("type", pa.int8(), nullable=False),
            pa.field("numRows", pa.int32(), nullable=False),
            pa.field("numCols", pa.int32(), nullable=False),
            pa.field("colPtrs", pa.list_(pa.int32())),
            pa.field("rowIndices", pa.list_(pa.int32())),
            pa.field("values", pa.list_(pa.float64())),
            pa.field("isTransposed", pa.bool_(), nullable=False),
        ]
    )


def _dense_vectors(rows: Sequence[np.ndarray]):
    """StructArray of N dense ml.linalg vectors (type=1, size/indices null)."""
    pa = _pa()
    n = len(rows)
    type_a = pa.array([1] * n, type=pa.int8())
    size_a = pa.nulls(n, type=pa.int32())
    idx_a = pa.nulls(n, type=pa.list_(pa.int32()))
    vals_a = pa.array([np.asarray(r, dtype=np.float64) for r in rows], type=pa.list_(pa.float64()))
    return pa.StructArray.from_arrays(
        [type_a, size_a, idx_a, vals_a], fields=list(_vector_struct_type())
    )


def _dense_matrices(mats: Sequence[np.ndarray]):
    """StructArray of dense ml.linalg matrices, row-major (isTransposed)."""
    pa = _pa()
    n = len(mats)
    type_a = pa.array([1] * n, type=pa.int8())
    nr = pa.array([int(m.shape[0]) for m in mats], type=pa.int32())
    nc = pa.array([int(m.shape[1]) for m in mats], type=pa.int32())
    colp = pa.nulls(n, type=pa.list_(pa.int32()))
    rowi = pa.nulls(n, type=pa.list_(pa.int32()))
    vals = pa.array(
        [np.ascontiguousarray(m, dtype=np.float64).ravel() for m in mats],
        type=pa.list_(pa.float64()),
    )
    tr = pa.array([True] * n, type=pa.bool_())
    return pa.StructArray.from_arrays(
        [type_a, nr, nc, colp, rowi, vals, tr], fields=list(_matrix_struct_type())
    )


def _read_vector(row: Dict[str, Any]) -> np.ndarray:
    """Decode one VectorUDT struct row (dense or sparse) to a dense array."""
    if int(row["type"]) == 1:
        return np.asarray(row["values"], dtype=np.float64)
    size = int(row["size"])
    out = np.zeros(size, dtype=np.float64)
    idx = np.asarray(row["indices"], dtype=np.int64)
    out[idx] = np.asarray(row["values"], dtype=np.float64)
    return out


def _read_matrix(row: Dict[str, Any]) -> np.ndarray:
    """Decode one MatrixUDT struct row to a dense [numRows, numCols] array."""
    nr, nc = int(row["numRows"]), int(row["numCols"])
    vals = np.asarray(row["values"], dtype=np.float64)
    if int(row["type"]) == 1:  # dense
        if row.get("isTransposed"):
            return vals.reshape(nr, nc)
        return vals.reshape(nc, nr).T
    # CSC sparse
    colp = np.asarray(row["colPtrs"], dtype=np.int64)
    rowi = np.asarray(row["rowIndices"], dtype=np.int64)
    out = np.zeros((nr, nc), dtype=np.float64)
    for c in range(nc):
        for p in range(colp[c], colp[c + 1]):
            out[rowi[p], c] = vals[p]
    if row.get("isTransposed"):
        return out.T
    return out


# ---------------------------------------------------------------------------
# directory plumbing
# ---------------------------------------------------------------------------


def _write_parquet_dir(dirpath: str, table, spark_schema: str) -> None:
    import pyarrow.parquet as pq

    os.makedirs(dirpath, exist_ok=True)
    table = table.replace_schema_metadata(
        {b"org.apache.spark.sql.parquet.row.metadata": spark_schema.encode()}
    )
    pq.write_table(table, os.path.join(dirpath, "part-00000.snappy.parquet"))
    open(os.path.join(dirpath, "_SUCCESS"), "w").close()


def _read_parquet_dir(dirpath: str):
    import pyarrow.parquet as pq

    files = sorted(
        os.path.join(dirpath, f)
        for f in os.listdir(dirpath)
        if f.endswith(".parquet")
    )
    if not files:
        raise FileNotFoundError(f"no parquet files under {dirpath}")
    import pyarrow as pa

    return pa.concat_tables([pq.read_table(f) for f in files], promote_options="permissive")


def _write_metadata(
    path: str,
    class_name: str,
    uid: str,
    param_map: Dict[str, Any],
    default_param_map: Dict[str, Any],
    extra: Optional[Dict[str, Any]] = None,
) -> None:
    meta: Dict[str, Any] = {
        "class": class_name,
        "timestamp": int(time.time() * 1000),
        "sparkVersion": SPARK_VERSION,
        "uid": uid,
        "paramMap": param_map,
        "defaultParamMap": default_param_map,
    }
    if extra:
        meta.update(extra)
    mdir = os.path.join(path, "metadata")
    os.makedirs(mdir, exist_ok=True)
    with open(os.path.join(mdir, "part-00000"), "w") as f:
        f.write(json.dumps(meta, separators=(",", ":")) + "\n")
    open(os.path.join(mdir, "_SUCCESS"), "w").close()


def read_spark_metadata(path: str) -> Dict[str, Any]:
    mdir = os.path.join(path, "metadata")
    parts = sorted(f for f in os.listdir(mdir) if f.startswith("part-"))
    if not parts:
        raise FileNotFoundError(f"no metadata part files under {mdir}")
    with open(os.path.join(mdir, parts[0])) as f:
        for line in f:
            line = line.strip()
            if line:
                return json.loads(line)
    raise ValueError(f"empty metadata file under {mdir}")


def spark_model_class(path: str) -> str:
    """The Spark class name a saved model directory declares."""
    return read_spark_metadata(path)["class"]


# ---------------------------------------------------------------------------
# param-map extraction (our Params -> Spark param names/values)
# ---------------------------------------------------------------------------

_JSON_SAFE = (str, int, float, bool)


def _collect_params(model, allowed: Sequence[str]) -> Tuple[Dict[str, Any], Dict[str, Any]]:
    """Split the model's params into (explicitly set, defaults), keeping only
    Spark-valid names with JSON-safe values."""
    set_map: Dict[str, Any] = {}
    def_map: Dict[str, Any] = {}
    for name in allowed:
        if not model.hasParam(name):
            continue
        p = model.getParam(name)
        if p in model._paramMap:
            v = model._paramMap[p]
            if isinstance(v, _JSON_SAFE):
                set_map[name] = v
        elif p in model._defaultParamMap:
            v = model._defaultParamMap[p]
            if isinstance(v, _JSON_SAFE):
                def_map[name] = v
    return set_map, def_map


# ---------------------------------------------------------------------------
# per-model writers
# ---------------------------------------------------------------------------

KMEANS_CLS = "org.apache.spark.ml.clustering.KMeansModel"
PCA_CLS = "org.apache.spark.ml.feature.PCAModel"
LINREG_CLS = "org.apache.spark.ml.regression.LinearRegressionModel"
LOGREG_CLS = "org.apache.spark.ml.classification.LogisticRegressionModel"
RFC_CLS = "org.apache.spark.ml.classification.RandomForestClassificationModel"
RFR_CLS = "org.apache.spark.ml.regression.RandomForestRegressionModel"
DTC_CLS = "org.apache.spark.ml.classification.DecisionTreeClassificationModel"
DTR_CLS = "org.apache.spark.ml.regression.DecisionTreeRegressionModel"


def _save_kmeans(model, path: str) -> None:
    pa = _pa()
    centers = np.asarray(model.cluster_centers_, dtype=np.float64)
    k = centers.shape[0]
    pm, dm = _collect_params(
        model,
        [
            "featuresCol", "predictionCol", "k", "initMode", "initSteps",
            "tol", "maxIter", "seed", "distanceMeasure",
        ],
    )
    dm.setdefault("distanceMeasure", "euclidean")
    _write_metadata(path, KMEANS_CLS, model.uid, pm, dm)
    table = pa.table(
        {
            "clusterIdx": pa.array(range(k), type=pa.int32()),
            "clusterCenter": _dense_vectors(list(centers)),
        }
    )
    schema = _schema_json(
        [_f("clusterIdx", "integer", False), _f("clusterCenter", VECTOR_UDT)]
    )
    _write_parquet_dir(os.path.join(path, "data"), table, schema)


def _save_pca(model, path: str) -> None:
    pa = _pa()
    # Spark stores pc as [d, k] (columns = principal components); ours is
    # components_ [k, d] rows = components
    comp = np.asarray(model.components_, dtype=np.float64)
    pc = comp.T  # [d, k]
    ev_ratio = np.asarray(model.explained_variance_ratio_, dtype=np.float64)
    pm, dm = _collect_params(model, ["inputCol", "outputCol", "k"])
    # our param names featuresCol/outputCol map onto Spark PCA inputCol/outputCol
    if model.hasParam("featuresCol") and "inputCol" not in pm:
        fc = model.getOrDefault("featuresCol")
        if isinstance(fc, str):
            pm["inputCol"] = fc
    if model.hasParam("k"):
        pm["k"] = int(model.getOrDefault("k"))
    if "outputCol" not in pm:
        oc = (
            model.getOrDefault("outputCol")
            if model.hasParam("outputCol")
            else "pca_features"
        )
        if isinstance(oc, str):
            pm["outputCol"] = oc
    _write_metadata(path, PCA_CLS, model.uid, pm, dm)
    table = pa.table(
        {
            "pc": _dense_matrices([pc]),
            "explainedVariance": _dense_vectors([ev_ratio]),
        }
    )
    schema = _schema_json([_f("pc", MATRIX_UDT), _f("explainedVariance", VECTOR_UDT)])
    _write_parquet_dir(os.path.join(path, "data"), table, schema)


def _save_linreg(model, path: str) -> None:
    pa = _pa()
    coef = np.asarray(model.coefficients, dtype=np.float64).ravel()
    pm, dm = _collect_params(
        model,
        [
            "featuresCol", "labelCol", "predictionCol", "regParam",
            "elasticNetParam", "maxIter", "tol", "fitIntercept",
            "standardization", "solver", "aggregationDepth", "loss", "epsilon",
        ],
    )
    dm.setdefault("loss", "squaredError")
    dm.setdefault("epsilon", 1.35)
    _write_metadata(path, LINREG_CLS, model.uid, pm, dm)
    table = pa.table(
        {
            "intercept": pa.array([float(model.intercept)], type=pa.float64()),
            "coefficients": _dense_vectors([coef]),
            "scale": pa.array([1.0], type=pa.float64()),
        }
    )
    schema = _schema_json(
        [
            _f("intercept", "double", False),
            _f("coefficients", VECTOR_UDT),
            _f("scale", "double", False),
        ]
    )
    _write_parquet_dir(os.path.join(path, "data"), table, schema)


def _save_logreg(model, path: str) -> None:
    pa = _pa()
    coef = np.atleast_2d(np.asarray(model._model_attributes["coef_"], dtype=np.float64))
    intercept = np.asarray(model._model_attributes["intercept_"], dtype=np.float64).ravel()
    n_classes = int(len(model._model_attributes["classes_"]))
    multinomial = coef.shape[0] > 1
    pm, dm = _collect_params(
        model,
        [
            "featuresCol", "labelCol", "predictionCol", "probabilityCol",
            "rawPredictionCol", "regParam", "elasticNetParam", "maxIter",
            "tol", "fitIntercept", "family", "standardization", "threshold",
            "aggregationDepth",
        ],
    )
    _write_metadata(path, LOGREG_CLS, model.uid, pm, dm)
    table = pa.table(
        {
            "numClasses": pa.array([n_classes], type=pa.int32()),
            "numFeatures": pa.array([int(coef.shape[1])], type=pa.int32()),
            "interceptVector": _dense_vectors([intercept]),
            "coefficientMatrix": _dense_matrices([coef]),
            "isMultinomial": pa.array([multinomial], type=pa.bool_()),
        }
    )
    schema = _schema_json(
        [
            _f("numClasses", "integer", False),
            _f("numFeatures", "integer", False),
            _f("interceptVector", VECTOR_UDT),
            _f("coefficientMatrix", MATRIX_UDT),
            _f("isMultinomial", "boolean", False),
        ]
    )
    _write_parquet_dir(os.path.join(path, "data"), table, schema)


# -- random forest ----------------------------------------------------------


def _tree_to_nodedata(
    t: Dict[str, np.ndarray], classification: bool
) -> List[Tuple[int, float, float, List[float], int, float, int, int, int, List[float], int]]:
    """Convert one of our flat trees to Spark NodeData rows with PRE-ORDER
    ids (Spark DecisionTreeModelReadWrite.NodeData.build numbering).

    Returns tuples: (id, prediction, impurity, impurityStats, rawCount,
    gain, leftChild, rightChild, splitFeature, splitThresholds, numCategories)
    """
    feature = t["feature"]
    threshold = t["threshold"]
    left = t["left"]
    right = t["right"]
    is_leaf = t["is_leaf"]
    gain = t["gain"]
    value = t["value"]

    out: List[Tuple] = []

    def stats_of(node: int) -> Tuple[float, float, List[float], int]:
        v = np.asarray(value[node], dtype=np.float64)
        if classification:
            cnt = float(v.sum())
            pred = float(np.argmax(v))
            p = v / max(cnt, 1e-300)
            impurity = float(1.0 - (p * p).sum())  # gini
            return pred, impurity, v.tolist(), int(round(cnt))
        mean, cnt = float(v[0]), float(v[1])
        # sum-of-squares not retained post-fit: stats reproduce the mean and
        # count exactly, node impurity reads as 0 (documented deviation)
        return mean, 0.0, [cnt, mean * cnt, mean * mean * cnt], int(round(cnt))

    # iterative pre-order with explicit new-id assignment
    next_id = [0]

    def build(node: int) -> int:
        my_id = next_id[0]
        next_id[0] += 1
        pred, imp, stats, raw = stats_of(node)
        if is_leaf[node]:
            out.append((my_id, pred, imp, stats, raw, -1.0, -1, -1, -1, [], -1))
            return my_id
        # placeholder, patched after children are numbered
        slot = len(out)
        out.append(None)  # type: ignore[arg-type]
        left_id = build(int(left[node]))
        right_id = build(int(right[node]))
        # Spark routes left on x <= thr; this engine on x < thr: step the
        # exported threshold down one ulp so both route identically
        thr = float(np.nextafter(np.float64(threshold[node]), -np.inf))
        out[slot] = (
            my_id, pred, imp, stats, raw, float(gain[node]),
            left_id, right_id, int(feature[node]), [thr], -1,
        )
        return my_id

    import sys

    old_limit = sys.getrecursionlimit()
    sys.setrecursionlimit(max(old_limit, 10000))
    try:
        build(0)
    finally:
        sys.setrecursionlimit(old_limit)
    return out


_NODEDATA_SPARK_SCHEMA = _schema_json(
    [
        _f("treeID", "integer", False),
        _f(
            "nodeData",
            {
                "type": "struct",
                "fields": [
                    {"name": "id", "type": "integer", "nullable": False, "metadata": {}},
                    {"name": "prediction", "type": "double", "nullable": False, "metadata": {}},
                    {"name": "impurity", "type": "double", "nullable": False, "metadata": {}},
                    {
                        "name": "impurityStats",
                        "type": {"type": "array", "elementType": "double", "containsNull": False},
                        "nullable": True,
                        "metadata": {},
                    },
                    {"name": "rawCount", "type": "long", "nullable": False, "metadata": {}},
                    {"name": "gain", "type": "double", "nullable": False, "metadata": {}},
                    {"name": "leftChild", "type": "integer", "nullable": False, "metadata": {}},
                    {"name": "rightChild", "type": "integer", "nullable": False, "metadata": {}},
                    {
                        "name": "split",
                        "type": {
                            "type": "struct",
                            "fields": [
                                {"name": "featureIndex", "type": "integer", "nullable": False, "metadata": {}},
                                {
                                    "name": "leftCategoriesOrThreshold",
                                    "type": {"type": "array", "elementType": "double", "containsNull": False},
                                    "nullable": True,
                                    "metadata": {},
                                },
                                {"name": "numCategories", "type": "integer", "nullable": False, "metadata": {}},
                            ],
                        },
                        "nullable": True,
                        "metadata": {},
                    },
                ],
            },
            True,
        ),
    ]
)


def _save_random_forest(model, path: str, classification: bool) -> None:
    pa = _pa()
    cls = RFC_CLS if classification else RFR_CLS
    tree_cls = DTC_CLS if classification else DTR_CLS
    allowed = [
        "featuresCol", "labelCol", "predictionCol", "numTrees", "maxDepth",
        "maxBins", "impurity", "subsamplingRate", "featureSubsetStrategy",
        "seed", "minInstancesPerNode", "minInfoGain", "bootstrap",
    ]
    if classification:
        allowed += ["probabilityCol", "rawPredictionCol"]
    pm, dm = _collect_params(model, allowed)
    extra: Dict[str, Any] = {
        "numFeatures": int(model.numFeatures),
        "numTrees": int(model.numTrees),
    }
    if classification:
        extra["numClasses"] = int(model.numClasses)
    _write_metadata(path, cls, model.uid, pm, dm, extra=extra)

    # treesMetadata: per-tree DefaultParamsWriter metadata JSON + weight
    tree_pm = {k: v for k, v in {**dm, **pm}.items() if k in (
        "featuresCol", "labelCol", "predictionCol", "probabilityCol",
        "rawPredictionCol", "maxDepth", "maxBins", "impurity", "seed",
        "minInstancesPerNode", "minInfoGain",
    )}
    trees = model.trees
    tm_rows = []
    for i in range(len(trees)):
        tj = {
            "class": tree_cls,
            "timestamp": int(time.time() * 1000),
            "sparkVersion": SPARK_VERSION,
            "uid": f"dtc_{model.uid}_{i}" if classification else f"dtr_{model.uid}_{i}",
            "paramMap": tree_pm,
            "defaultParamMap": {},
        }
        tm_rows.append((i, json.dumps(tj, separators=(",", ":")), 1.0))
    tm_table = pa.table(
        {
            "treeID": pa.array([r[0] for r in tm_rows], type=pa.int32()),
            "metadata": pa.array([r[1] for r in tm_rows], type=pa.string()),
            "weights": pa.array([r[2] for r in tm_rows], type=pa.float64()),
        }
    )
    tm_schema = _schema_json(
        [_f("treeID", "integer", False), _f("metadata", "string"), _f("weights", "double", False)]
    )
    _write_parquet_dir(os.path.join(path, "treesMetadata"), tm_table, tm_schema)

    # data: EnsembleNodeData(treeID, nodeData)
    tree_ids: List[int] = []
    node_rows: List[Dict[str, Any]] = []
    for tid, t in enumerate(trees):
        for (
            nid, pred, imp, stats, raw, g, lc, rc, sf, sthr, ncat
        ) in _tree_to_nodedata(t, classification):
            tree_ids.append(tid)
            node_rows.append(
                {
                    "id": nid,
                    "prediction": pred,
                    "impurity": imp,
                    "impurityStats": stats,
                    "rawCount": raw,
                    "gain": g,
                    "leftChild": lc,
                    "rightChild": rc,
                    "split": {
                        "featureIndex": sf,
                        "leftCategoriesOrThreshold": sthr,
                        "numCategories": ncat,
                    },
                }
            )
    node_t = pa.struct(
        [
            pa.field("id", pa.int32(), nullable=False),
            pa.field("prediction", pa.float64(), nullable=False),
            pa.field("impurity", pa.float64(), nullable=False),
            pa.field("impurityStats", pa.list_(pa.float64())),
            pa.field("rawCount", pa.int64(), nullable=False),
            pa.field("gain", pa.float64(), nullable=False),
            pa.field("leftChild", pa.int32(), nullable=False),
            pa.field("rightChild", pa.int32(), nullable=False),
            pa.field(
                "split",
                pa.struct(
                    [
                        pa.field("featureIndex", pa.int32(), nullable=False),
                        pa.field("leftCategoriesOrThreshold", pa.list_(pa.float64())),
                        pa.field("numCategories", pa.int32(), nullable=False),
                    ]
                ),
            ),
        ]
    )
    data_table = pa.table(
        {
            "treeID": pa.array(tree_ids, type=pa.int32()),
            "nodeData": pa.array(node_rows, type=node_t),
        }
    )
    _write_parquet_dir(os.path.join(path, "data"), data_table, _NODEDATA_SPARK_SCHEMA)


# ---------------------------------------------------------------------------
# public save/load
# ---------------------------------------------------------------------------


def save_spark_model(model, path: str, overwrite: bool = False) -> None:
    """Write `model` to `path` in stock Spark ML's persistence format so
    `pyspark.ml.<family>.<Model>.load(path)` reads it (reference capability:
    `cpu()` + Spark writers, reference core.py:268-355)."""
    from ..models.clustering import KMeansModel
    from ..models.feature import PCAModel
    from ..models.regression import LinearRegressionModel, RandomForestRegressionModel
    from ..models.classification import (
        LogisticRegressionModel,
        RandomForestClassificationModel,
    )

    if os.path.exists(path):
        if not overwrite:
            raise FileExistsError(f"{path} exists (use overwrite=True)")
        import shutil

        shutil.rmtree(path)
    os.makedirs(path, exist_ok=True)

    if isinstance(model, KMeansModel):
        _save_kmeans(model, path)
    elif isinstance(model, PCAModel):
        _save_pca(model, path)
    elif isinstance(model, LogisticRegressionModel):
        _save_logreg(model, path)
    elif isinstance(model, RandomForestClassificationModel):
        _save_random_forest(model, path, classification=True)
    elif isinstance(model, RandomForestRegressionModel):
        _save_random_forest(model, path, classification=False)
    elif isinstance(model, LinearRegressionModel):
        _save_linreg(model, path)
    else:
        raise TypeError(
            f"no Spark persistence mapping for {type(model).__name__} "
            "(supported: KMeansModel, PCAModel, LinearRegressionModel, "
            "LogisticRegressionModel, RandomForest*Model)"
        )


def _apply_params(model, meta: Dict[str, Any]) -> None:
    """Set whatever metadata params our model also has (name-compatible)."""
    for src in ("defaultParamMap", "paramMap"):
        for name, v in (meta.get(src) or {}).items():
            target = name
            if not model.hasParam(target):
                # Spark PCA uses inputCol/outputCol; ours uses featuresCol
                if name == "inputCol" and model.hasParam("featuresCol"):
                    target = "featuresCol"
                else:
                    continue
            try:
                model._set(**{target: v})
            except Exception:
                pass


def _load_kmeans(path: str, meta: Dict[str, Any]):
    from ..models.clustering import KMeansModel

    tbl = _read_parquet_dir(os.path.join(path, "data"))
    rows = tbl.to_pylist()
    rows.sort(key=lambda r: r["clusterIdx"])
    centers = np.stack([_read_vector(r["clusterCenter"]) for r in rows])
    m = KMeansModel(cluster_centers_=centers)
    _apply_params(m, meta)
    return m


def _load_pca(path: str, meta: Dict[str, Any]):
    from ..models.feature import PCAModel

    tbl = _read_parquet_dir(os.path.join(path, "data"))
    row = tbl.to_pylist()[0]
    pc = _read_matrix(row["pc"])  # [d, k]
    evr = _read_vector(row["explainedVariance"])
    comp = pc.T  # ours: [k, d]
    m = PCAModel(
        components_=comp,
        explained_variance_=evr,  # ratios stand in: Spark persists only ratios
        explained_variance_ratio_=evr,
        mean_=np.zeros(comp.shape[1]),
    )
    _apply_params(m, meta)
    return m


def _load_linreg(path: str, meta: Dict[str, Any]):
    from ..models.regression import LinearRegressionModel

    tbl = _read_parquet_dir(os.path.join(path, "data"))
    row = tbl.to_pylist()[0]
    m = LinearRegressionModel(
        coef_=_read_vector(row["coefficients"]), intercept_=float(row["intercept"])
    )
    _apply_params(m, meta)
    return m


def _load_logreg(path: str, meta: Dict[str, Any]):
    from ..models.classification import LogisticRegressionModel

    tbl = _read_parquet_dir(os.path.join(path, "data"))
    row = tbl.to_pylist()[0]
    coef = _read_matrix(row["coefficientMatrix"])
    intercept = _read_vector(row["interceptVector"])
    n_classes = int(row["numClasses"])
    m = LogisticRegressionModel(
        coef_=coef,
        intercept_=intercept,
        classes_=np.arange(n_classes, dtype=np.float64),
    )
    _apply_params(m, meta)
    return m


def _load_random_forest(path: str, meta: Dict[str, Any], classification: bool):
    import pickle

    from ..models.classification import RandomForestClassificationModel
    from ..models.regression import RandomForestRegressionModel

    tbl = _read_parquet_dir(os.path.join(path, "data"))
    by_tree: Dict[int, List[Dict[str, Any]]] = {}
    for r in tbl.to_pylist():
        by_tree.setdefault(int(r["treeID"]), []).append(r["nodeData"])

    n_classes = int(meta.get("numClasses", 0))
    trees: List[Dict[str, np.ndarray]] = []
    for tid in sorted(by_tree):
        nodes = sorted(by_tree[tid], key=lambda nd: nd["id"])
        n = len(nodes)
        feature = np.full(n, -1, dtype=np.int32)
        threshold = np.zeros(n, dtype=np.float32)
        left = np.full(n, -1, dtype=np.int32)
        right = np.full(n, -1, dtype=np.int32)
        is_leaf = np.ones(n, dtype=bool)
        gain = np.zeros(n, dtype=np.float32)
        width = n_classes if classification else 2
        value = np.zeros((n, width), dtype=np.float32)
        for nd in nodes:
            i = int(nd["id"])
            stats = np.asarray(nd["impurityStats"] or [], dtype=np.float64)
            if classification:
                if len(stats) == width:
                    value[i] = stats
                else:
                    value[i, int(nd["prediction"])] = max(float(nd["rawCount"]), 1.0)
            else:
                cnt = float(stats[0]) if len(stats) else float(nd["rawCount"])
                value[i] = (float(nd["prediction"]), cnt)
            if int(nd["leftChild"]) >= 0:
                is_leaf[i] = False
                left[i] = int(nd["leftChild"])
                right[i] = int(nd["rightChild"])
                gain[i] = float(nd["gain"])
                feature[i] = int(nd["split"]["featureIndex"])
                thr_list = nd["split"]["leftCategoriesOrThreshold"]
                # inverse of the export step: Spark <= thr == ours < nextafter(thr, +inf)
                threshold[i] = np.float32(
                    np.nextafter(np.float64(thr_list[0]), np.inf)
                ) if thr_list else 0.0
        trees.append(
            {
                "feature": feature,
                "threshold": threshold,
                "left": left,
                "right": right,
                "is_leaf": is_leaf,
                "gain": gain,
                "value": value,
            }
        )

    forest_blob = np.frombuffer(pickle.dumps(trees), dtype=np.uint8)
    n_features = int(meta.get("numFeatures", 0))
    cls = RandomForestClassificationModel if classification else RandomForestRegressionModel
    m = cls(
        forest_=forest_blob,
        n_classes_=n_classes,
        n_features_=n_features,
        n_trees_=len(trees),
    )
    _apply_params(m, meta)
    return m


_LOADERS = {
    KMEANS_CLS: _load_kmeans,
    PCA_CLS: _load_pca,
    LINREG_CLS: _load_linreg,
    LOGREG_CLS: _load_logreg,
}


def load_spark_model(path: str):
    """Load a Spark-ML-format model directory (written by stock pyspark OR by
    save_spark_model) into the corresponding model of this package."""
    meta = read_spark_metadata(path)
    cls = meta["class"]
    if cls in _LOADERS:
        return _LOADERS[cls](path, meta)
    if cls == RFC_CLS:
        return _load_random_forest(path, meta, classification=True)
    if cls == RFR_CLS:
        return _load_random_forest(path, meta, classification=False)
    raise TypeError(f"no loader for Spark model class {cls!r}")
