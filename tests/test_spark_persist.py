"""Spark-ML on-disk model format: structural schema checks + round-trips.

Stock pyspark is not installable in this environment, so loadability is
verified structurally: the directory layout, metadata JSON keys, parquet
footer Spark-schema metadata (UDT declarations), and the exact VectorUDT /
MatrixUDT struct encodings are asserted against Spark's documented format
(org.apache.spark.ml.linalg UDTs; DefaultParamsWriter metadata), and every
model round-trips through load_spark_model with identical predictions.
"""

import json
import os

import numpy as np
import pytest
from sklearn.datasets import make_blobs, make_classification, make_regression

from spark_rapids_ml_amd import (
    KMeans,
    LinearRegression,
    LogisticRegression,
    PCA,
    RandomForestClassifier,
    RandomForestRegressor,
)
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.spark import load_spark_model, save_spark_model, spark_model_class


def _footer_schema(parquet_dir):
    import pyarrow.parquet as pq

    files = [f for f in os.listdir(parquet_dir) if f.endswith(".parquet")]
    assert files, f"no parquet part files in {parquet_dir}"
    md = pq.ParquetFile(os.path.join(parquet_dir, files[0])).schema_arrow.metadata
    assert b"org.apache.spark.sql.parquet.row.metadata" in md, (
        "Spark schema JSON missing from parquet footer - stock Spark cannot "
        "restore UDT columns without it"
    )
    return json.loads(md[b"org.apache.spark.sql.parquet.row.metadata"])


def _meta(path):
    with open(os.path.join(path, "metadata", "part-00000")) as f:
        return json.loads(f.readline())


def _check_common_layout(path, expected_class):
    assert os.path.exists(os.path.join(path, "metadata", "part-00000"))
    assert os.path.exists(os.path.join(path, "metadata", "_SUCCESS"))
    assert os.path.exists(os.path.join(path, "data", "_SUCCESS"))
    meta = _meta(path)
    for key in ("class", "timestamp", "sparkVersion", "uid", "paramMap", "defaultParamMap"):
        assert key in meta, f"metadata missing {key}"
    assert meta["class"] == expected_class
    assert spark_model_class(path) == expected_class
    return meta


def test_kmeans_spark_format(tmp_path):
    X, _ = make_blobs(n_samples=300, n_features=8, centers=4, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float32))
    m = KMeans(k=4, maxIter=10, seed=1).fit(df)
    p = str(tmp_path / "km")
    m.saveAsSparkModel(p)

    meta = _check_common_layout(p, "org.apache.spark.ml.clustering.KMeansModel")
    assert meta["paramMap"]["k"] == 4

    schema = _footer_schema(os.path.join(p, "data"))
    names = [f["name"] for f in schema["fields"]]
    assert names == ["clusterIdx", "clusterCenter"]
    udt = schema["fields"][1]["type"]
    assert udt["type"] == "udt"
    assert udt["class"] == "org.apache.spark.ml.linalg.VectorUDT"
    assert udt["pyClass"] == "pyspark.ml.linalg.VectorUDT"
    sql_fields = [f["name"] for f in udt["sqlType"]["fields"]]
    assert sql_fields == ["type", "size", "indices", "values"]

    # physical rows: dense encoding type=1, values only
    import pyarrow.parquet as pq

    tbl = pq.read_table(os.path.join(p, "data"))
    rows = sorted(tbl.to_pylist(), key=lambda r: r["clusterIdx"])
    assert [r["clusterIdx"] for r in rows] == [0, 1, 2, 3]
    for r in rows:
        assert r["clusterCenter"]["type"] == 1
        assert r["clusterCenter"]["size"] is None
        assert r["clusterCenter"]["indices"] is None
        assert len(r["clusterCenter"]["values"]) == 8
    got = np.stack([r["clusterCenter"]["values"] for r in rows])
    np.testing.assert_allclose(got, np.asarray(m.cluster_centers_, np.float64), rtol=1e-6)

    # round-trip
    m2 = load_spark_model(p)
    np.testing.assert_allclose(m2.cluster_centers_, m.cluster_centers_, rtol=1e-6)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_array_equal(pred1, pred2)


def test_pca_spark_format(tmp_path):
    rng = np.random.default_rng(0)
    X = rng.normal(size=(200, 10))
    m = PCA(k=3).fit(DataFrame.from_numpy(X))
    p = str(tmp_path / "pca")
    m.saveAsSparkModel(p)

    _check_common_layout(p, "org.apache.spark.ml.feature.PCAModel")
    schema = _footer_schema(os.path.join(p, "data"))
    names = [f["name"] for f in schema["fields"]]
    assert names == ["pc", "explainedVariance"]
    assert schema["fields"][0]["type"]["class"] == "org.apache.spark.ml.linalg.MatrixUDT"
    mat_fields = [f["name"] for f in schema["fields"][0]["type"]["sqlType"]["fields"]]
    assert mat_fields == [
        "type", "numRows", "numCols", "colPtrs", "rowIndices", "values", "isTransposed",
    ]

    import pyarrow.parquet as pq

    row = pq.read_table(os.path.join(p, "data")).to_pylist()[0]
    # Spark stores pc as [numFeatures, k]
    assert row["pc"]["numRows"] == 10
    assert row["pc"]["numCols"] == 3
    assert row["pc"]["type"] == 1

    m2 = load_spark_model(p)
    np.testing.assert_allclose(m2.components_, m.components_, rtol=1e-6)
    np.testing.assert_allclose(
        m2.explained_variance_ratio_, m.explained_variance_ratio_, rtol=1e-6
    )


def test_linreg_spark_format(tmp_path):
    X, y = make_regression(n_samples=300, n_features=6, noise=1.0, random_state=0)
    m = LinearRegression().fit(DataFrame.from_numpy(X, y))
    p = str(tmp_path / "lr")
    m.saveAsSparkModel(p)

    _check_common_layout(p, "org.apache.spark.ml.regression.LinearRegressionModel")
    schema = _footer_schema(os.path.join(p, "data"))
    names = [f["name"] for f in schema["fields"]]
    assert names == ["intercept", "coefficients", "scale"]

    import pyarrow.parquet as pq

    row = pq.read_table(os.path.join(p, "data")).to_pylist()[0]
    assert row["scale"] == 1.0
    np.testing.assert_allclose(
        row["coefficients"]["values"], np.asarray(m.coefficients, np.float64), rtol=1e-6
    )

    m2 = load_spark_model(p)
    np.testing.assert_allclose(m2.coefficients, m.coefficients, rtol=1e-6)
    assert np.isclose(m2.intercept, m.intercept)


@pytest.mark.parametrize("family", ["binomial", "multinomial"])
def test_logreg_spark_format(tmp_path, family):
    n_cls = 2 if family == "binomial" else 3
    X, y = make_classification(
        n_samples=400, n_features=8, n_informative=5, n_classes=n_cls,
        n_clusters_per_class=1, random_state=0,
    )
    df = DataFrame.from_numpy(X, y.astype(np.float64))
    m = LogisticRegression(maxIter=50, family=family).fit(df)
    p = str(tmp_path / "logreg")
    m.saveAsSparkModel(p)

    _check_common_layout(p, "org.apache.spark.ml.classification.LogisticRegressionModel")
    schema = _footer_schema(os.path.join(p, "data"))
    names = [f["name"] for f in schema["fields"]]
    assert names == [
        "numClasses", "numFeatures", "interceptVector", "coefficientMatrix", "isMultinomial",
    ]

    import pyarrow.parquet as pq

    row = pq.read_table(os.path.join(p, "data")).to_pylist()[0]
    assert row["numClasses"] == n_cls
    assert row["numFeatures"] == 8
    assert row["isMultinomial"] == (family == "multinomial")
    cm = row["coefficientMatrix"]
    assert cm["isTransposed"] is True  # row-major export

    m2 = load_spark_model(p)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_array_equal(pred1, pred2)


def test_rfc_spark_format(tmp_path):
    X, y = make_classification(
        n_samples=500, n_features=8, n_informative=5, n_classes=3,
        n_clusters_per_class=1, random_state=0,
    )
    df = DataFrame.from_numpy(X.astype(np.float32), y.astype(np.float64))
    m = RandomForestClassifier(numTrees=5, maxDepth=5, seed=1).fit(df)
    p = str(tmp_path / "rfc")
    m.saveAsSparkModel(p)

    meta = _check_common_layout(
        p, "org.apache.spark.ml.classification.RandomForestClassificationModel"
    )
    # ensemble extras live at metadata top level (EnsembleModelReadWrite)
    assert meta["numFeatures"] == 8
    assert meta["numClasses"] == 3
    assert meta["numTrees"] == 5

    # treesMetadata: one row per tree, parseable per-tree metadata JSON
    import pyarrow.parquet as pq

    tm = pq.read_table(os.path.join(p, "treesMetadata")).to_pylist()
    assert len(tm) == 5
    assert sorted(r["treeID"] for r in tm) == [0, 1, 2, 3, 4]
    tj = json.loads(tm[0]["metadata"])
    assert tj["class"] == "org.apache.spark.ml.classification.DecisionTreeClassificationModel"
    assert tm[0]["weights"] == 1.0

    # nodeData: pre-order ids starting at 0 per tree; leaves have -1 markers
    nd = pq.read_table(os.path.join(p, "data")).to_pylist()
    by_tree = {}
    for r in nd:
        by_tree.setdefault(r["treeID"], []).append(r["nodeData"])
    for tid, nodes in by_tree.items():
        ids = sorted(n["id"] for n in nodes)
        assert ids == list(range(len(nodes))), "node ids must be 0..n-1"
        for n in nodes:
            if n["leftChild"] == -1:
                assert n["rightChild"] == -1
                assert n["gain"] == -1.0
                assert n["split"]["featureIndex"] == -1
                assert n["split"]["leftCategoriesOrThreshold"] == []
            else:
                assert n["split"]["numCategories"] == -1
                assert len(n["split"]["leftCategoriesOrThreshold"]) == 1
            assert len(n["impurityStats"]) == 3  # class counts
            assert n["rawCount"] >= 1

    m2 = load_spark_model(p)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_array_equal(pred1, pred2)


def test_rfr_spark_format(tmp_path):
    X, y = make_regression(n_samples=500, n_features=8, noise=2.0, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float32), y)
    m = RandomForestRegressor(numTrees=4, maxDepth=5, seed=1).fit(df)
    p = str(tmp_path / "rfr")
    m.saveAsSparkModel(p)

    meta = _check_common_layout(
        p, "org.apache.spark.ml.regression.RandomForestRegressionModel"
    )
    assert meta["numFeatures"] == 8
    assert meta["numTrees"] == 4
    assert "numClasses" not in meta

    m2 = load_spark_model(p)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_allclose(pred1, pred2, rtol=1e-5)


def test_threshold_semantics_preserved_across_spark_roundtrip(tmp_path):
    """Integer-valued data where split thresholds land exactly on data
    values: the < vs <= convention translation (nextafter at export, inverse
    at import) must keep predictions bit-identical."""
    rng = np.random.default_rng(0)
    X = rng.integers(0, 10, size=(600, 4)).astype(np.float32)
    y = (X[:, 0] >= 5).astype(np.float64)
    df = DataFrame.from_numpy(X, y)
    m = RandomForestClassifier(
        numTrees=1, maxDepth=4, bootstrap=False, seed=3, featureSubsetStrategy="all"
    ).fit(df)
    p = str(tmp_path / "rf_thr")
    m.saveAsSparkModel(p)
    m2 = load_spark_model(p)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_array_equal(pred1, pred2)
    assert (pred1 == y).all()


def test_overwrite_semantics(tmp_path):
    X, _ = make_blobs(n_samples=100, n_features=4, centers=2, random_state=0)
    df = DataFrame.from_numpy(X.astype(np.float32))
    m = KMeans(k=2, maxIter=5, seed=1).fit(df)
    p = str(tmp_path / "km")
    m.saveAsSparkModel(p)
    with pytest.raises(FileExistsError):
        m.saveAsSparkModel(p)
    m.saveAsSparkModel(p, overwrite=True)
    assert spark_model_class(p).endswith("KMeansModel")


def test_load_sparse_vector_model_written_by_stock_spark(tmp_path):
    """Stock Spark may persist SPARSE vectors/matrices (type=0). Hand-craft
    such a directory (as Spark would write it) and load it."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from spark_rapids_ml_amd.spark.persist import (
        LINREG_CLS,
        _vector_struct_type,
        _write_metadata,
    )

    p = str(tmp_path / "sparse_lr")
    os.makedirs(os.path.join(p, "data"))
    _write_metadata(p, LINREG_CLS, "LinearRegression_stock", {}, {})

    # sparse coefficients: size=6, indices [1, 4], values [2.0, -3.0]
    sparse_vec = pa.StructArray.from_arrays(
        [
            pa.array([0], type=pa.int8()),
            pa.array([6], type=pa.int32()),
            pa.array([[1, 4]], type=pa.list_(pa.int32())),
            pa.array([[2.0, -3.0]], type=pa.list_(pa.float64())),
        ],
        fields=list(_vector_struct_type()),
    )
    table = pa.table(
        {
            "intercept": pa.array([0.5], type=pa.float64()),
            "coefficients": sparse_vec,
            "scale": pa.array([1.0], type=pa.float64()),
        }
    )
    pq.write_table(table, os.path.join(p, "data", "part-00000.snappy.parquet"))

    m = load_spark_model(p)
    np.testing.assert_allclose(
        np.asarray(m.coefficients), [0.0, 2.0, 0.0, 0.0, -3.0, 0.0]
    )
    assert np.isclose(m.intercept, 0.5)


@pytest.mark.parametrize("shape", [(60, 1), (60, 2), (200, 33)])
def test_spark_format_edge_shapes(tmp_path, shape):
    """Tiny/odd shapes must survive the Spark-format round trip."""
    n, d = shape
    rng = np.random.default_rng(0)
    X = rng.normal(size=(n, d)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)

    km = KMeans(k=2, maxIter=5, seed=1).fit(DataFrame.from_numpy(X))
    p1 = str(tmp_path / "km")
    km.saveAsSparkModel(p1)
    km2 = load_spark_model(p1)
    np.testing.assert_allclose(km2.cluster_centers_, km.cluster_centers_, rtol=1e-6)

    lr = LinearRegression().fit(DataFrame.from_numpy(X, y))
    p2 = str(tmp_path / "lr")
    lr.saveAsSparkModel(p2)
    lr2 = load_spark_model(p2)
    np.testing.assert_allclose(
        np.asarray(lr2.coefficients), np.asarray(lr.coefficients), rtol=1e-5, atol=1e-7
    )


def test_spark_format_deep_forest_roundtrip(tmp_path):
    """A deeper forest (many nodes, both tasks) round-trips bit-equal
    predictions through the Spark node-data format."""
    from sklearn.datasets import make_classification

    X, y = make_classification(n_samples=2000, n_features=10, n_informative=6,
                               n_classes=4, n_clusters_per_class=1, random_state=1)
    df = DataFrame.from_numpy(X.astype(np.float32), y.astype(np.float64))
    m = RandomForestClassifier(numTrees=9, maxDepth=9, seed=2).fit(df)
    p = str(tmp_path / "deep")
    m.saveAsSparkModel(p)
    m2 = load_spark_model(p)
    pred1 = np.asarray(m.transform(df)[m.getOrDefault("predictionCol")])
    pred2 = np.asarray(m2.transform(df)[m2.getOrDefault("predictionCol")])
    np.testing.assert_array_equal(pred1, pred2)
    # probability parity too (vote normalization must survive)
    pr1 = np.asarray(m.transform(df)[m.getOrDefault("probabilityCol")])
    pr2 = np.asarray(m2.transform(df)[m2.getOrDefault("probabilityCol")])
    np.testing.assert_allclose(pr1, pr2, atol=1e-6)
