"""pyspark bridge tests against a stub pyspark (pyspark itself is not
installable in this environment — no network; the stub reproduces the exact
API surface the bridge touches: DataFrame.toArrow/schema/columns/select,
ml.functions.vector_to_array, sparkSession.createDataFrame)."""

import sys
import types

import numpy as np
import pytest
from sklearn.datasets import make_blobs

from spark_rapids_ml_amd import KMeans, LinearRegression
from spark_rapids_ml_amd.data import DataFrame as LocalDF

from .dist_utils import run_distributed


# ---------------------------------------------------------------------------
# stub pyspark
# ---------------------------------------------------------------------------


def _install_stub_pyspark():
    """Build pyspark/pyspark.sql/pyspark.ml stub modules in sys.modules.
    Returns the StubDataFrame class."""
    import pyarrow as pa

    pyspark = types.ModuleType("pyspark")
    sql = types.ModuleType("pyspark.sql")
    sql_df_mod = types.ModuleType("pyspark.sql.dataframe")
    sql_functions = types.ModuleType("pyspark.sql.functions")
    ml = types.ModuleType("pyspark.ml")
    ml_functions = types.ModuleType("pyspark.ml.functions")

    class _Field:
        def __init__(self, name, type_name):
            self.name = name
            self.dataType = type("VectorUDT" if type_name == "vector" else "ArrayType", (), {})()

    class _Schema:
        def __init__(self, fields):
            self.fields = fields

    class _Col:
        def __init__(self, name):
            self.name = name
            self.unwrapped = False

        def alias(self, name):
            self.name = name
            return self

    class StubSparkSession:
        def createDataFrame(self, pdf):
            table = pa.Table.from_pandas(pdf)
            return StubDataFrame(table)

    class StubDataFrame:
        """Mimics pyspark.sql.DataFrame for the bridge call sequence."""

        def __init__(self, table: pa.Table, vector_cols=()):
            self._table = table
            self._vector_cols = set(vector_cols)
            self.sparkSession = StubSparkSession()

        @property
        def columns(self):
            return list(self._table.column_names)

        @property
        def schema(self):
            return _Schema(
                [
                    _Field(n, "vector" if n in self._vector_cols else "array")
                    for n in self._table.column_names
                ]
            )

        def select(self, *cols):
            # vector_to_array(F.col(c)) unwraps in stock Spark; the stub's
            # "vectors" are already list arrays, so select is a projection
            names = [c.name for c in cols]
            return StubDataFrame(self._table.select(names))

        def toArrow(self):
            return self._table

    # modules must claim pyspark.* so is_pyspark_dataframe sees them
    StubDataFrame.__module__ = "pyspark.sql.dataframe"
    StubDataFrame.__name__ = "DataFrame"
    StubDataFrame.__qualname__ = "DataFrame"

    def vector_to_array(col, dtype="float64"):
        col.unwrapped = True
        return col

    def F_col(name):
        return _Col(name)

    sql_functions.col = F_col
    ml_functions.vector_to_array = vector_to_array
    sql_df_mod.DataFrame = StubDataFrame
    sql.DataFrame = StubDataFrame
    sql.functions = sql_functions
    pyspark.sql = sql
    pyspark.ml = ml
    ml.functions = ml_functions

    # real-ish pyspark.ml submodules for the proxy test
    for sub in ("feature", "clustering", "classification", "regression", "tuning"):
        m = types.ModuleType(f"pyspark.ml.{sub}")
        # a stock class that must keep resolving through the proxy
        class _CpuOnly:  # noqa: N801
            pass

        _CpuOnly.__name__ = f"CpuOnly_{sub}"
        m.CpuOnlyThing = _CpuOnly
        if sub == "clustering":
            m.KMeans = type("KMeans", (), {"_stock": True})
            m.BisectingKMeans = type("BisectingKMeans", (), {})
        sys.modules[f"pyspark.ml.{sub}"] = m
        setattr(ml, sub, m)

    sys.modules["pyspark"] = pyspark
    sys.modules["pyspark.sql"] = sql
    sys.modules["pyspark.sql.dataframe"] = sql_df_mod
    sys.modules["pyspark.sql.functions"] = sql_functions
    sys.modules["pyspark.ml"] = ml
    sys.modules["pyspark.ml.functions"] = ml_functions
    return StubDataFrame


def _remove_stub_pyspark():
    for name in [n for n in sys.modules if n == "pyspark" or n.startswith("pyspark.")]:
        del sys.modules[name]


@pytest.fixture()
def stub_pyspark():
    cls = _install_stub_pyspark()
    try:
        yield cls
    finally:
        _remove_stub_pyspark()


def _blob_table(n=300, d=6, vector_col=False):
    import pyarrow as pa

    X, _ = make_blobs(n_samples=n, n_features=d, centers=3, random_state=0)
    X = X.astype(np.float64)
    feats = pa.FixedSizeListArray.from_arrays(pa.array(X.ravel()), d)
    return pa.table({"features": feats}), X


# ---------------------------------------------------------------------------
# detection + conversion
# ---------------------------------------------------------------------------


def test_is_pyspark_dataframe(stub_pyspark):
    from spark_rapids_ml_amd.spark.bridge import is_pyspark_dataframe

    table, _ = _blob_table()
    assert is_pyspark_dataframe(stub_pyspark(table))
    assert not is_pyspark_dataframe(LocalDF({"x": np.zeros(3)}))
    assert not is_pyspark_dataframe("not a df")


def test_fit_on_stub_spark_dataframe(stub_pyspark):
    table, X = _blob_table()
    sdf = stub_pyspark(table)
    m_spark = KMeans(k=3, maxIter=10, seed=1).fit(sdf)
    m_local = KMeans(k=3, maxIter=10, seed=1).fit(LocalDF.from_numpy(X.astype(np.float32)))
    np.testing.assert_allclose(
        np.sort(np.asarray(m_spark.cluster_centers_), axis=0),
        np.sort(np.asarray(m_local.cluster_centers_), axis=0),
        rtol=1e-5,
    )


def test_transform_returns_spark_dataframe(stub_pyspark):
    table, X = _blob_table()
    sdf = stub_pyspark(table)
    m = KMeans(k=3, maxIter=10, seed=1).fit(sdf)
    out = m.transform(sdf)
    # bridge must hand back a pyspark (stub) DataFrame on the same session
    assert type(out).__module__.startswith("pyspark.")
    assert "prediction" in out.columns
    pred = np.asarray(out.toArrow().column("prediction"))
    assert pred.shape[0] == X.shape[0]
    assert set(np.unique(pred)) <= {0, 1, 2}


def test_vector_udt_unwrap_path(stub_pyspark):
    """A VectorUDT features column must route through vector_to_array before
    Arrow collection (Arrow cannot ship UDTs; reference core.py:220-265)."""
    from spark_rapids_ml_amd.spark.bridge import spark_to_local

    table, X = _blob_table()
    sdf = stub_pyspark(table, vector_cols=["features"])
    local = spark_to_local(sdf)
    np.testing.assert_allclose(np.asarray(local["features"]), X)


def test_linreg_fit_on_stub_spark(stub_pyspark):
    import pyarrow as pa

    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 5))
    w = rng.normal(size=5)
    y = X @ w + 0.3
    feats = pa.FixedSizeListArray.from_arrays(pa.array(X.ravel()), 5)
    table = pa.table({"features": feats, "label": pa.array(y)})
    m = LinearRegression().fit(stub_pyspark(table))
    np.testing.assert_allclose(np.asarray(m.coefficients), w, atol=1e-6)
    assert np.isclose(m.intercept, 0.3, atol=1e-6)


# ---------------------------------------------------------------------------
# pyspark.ml proxying
# ---------------------------------------------------------------------------


def test_accelerate_pyspark_proxies_modules(stub_pyspark):
    import importlib

    from spark_rapids_ml_amd import KMeans as OurKMeans
    from spark_rapids_ml_amd.install import accelerate_pyspark

    proxied = accelerate_pyspark()
    assert "pyspark.ml.clustering" in proxied

    clustering = importlib.import_module("pyspark.ml.clustering")
    # accelerated name resolves to ours
    assert clustering.KMeans is OurKMeans
    # non-accelerated names keep resolving to stock pyspark
    assert clustering.BisectingKMeans.__name__ == "BisectingKMeans"
    assert not hasattr(clustering.BisectingKMeans, "_fit_array")
    assert clustering.CpuOnlyThing.__name__ == "CpuOnly_clustering"

    # idempotent
    assert accelerate_pyspark() == proxied


def test_accelerate_pyspark_without_pyspark():
    from spark_rapids_ml_amd.install import accelerate_pyspark

    _remove_stub_pyspark()
    with pytest.raises(ImportError):
        accelerate_pyspark()


# ---------------------------------------------------------------------------
# distributed sharding of the Arrow export
# ---------------------------------------------------------------------------


def _dist_spark_fit(payload):
    import pickle

    import pyarrow as pa

    from tests.test_spark_bridge import _install_stub_pyspark, _remove_stub_pyspark
    from spark_rapids_ml_amd.spark.bridge import spark_to_local

    X = pickle.loads(payload)
    d = X.shape[1]
    StubDataFrame = _install_stub_pyspark()
    try:
        # many small batches so round-robin sharding has units to deal
        feats_chunks = []
        bs = 32
        for s in range(0, len(X), bs):
            chunk = X[s : s + bs]
            feats_chunks.append(
                pa.record_batch(
                    {"features": pa.FixedSizeListArray.from_arrays(pa.array(chunk.ravel()), d)}
                )
            )
        table = pa.Table.from_batches(feats_chunks)
        sdf = StubDataFrame(table)
        local = spark_to_local(sdf, shard=True)
        rows = np.asarray(local["features"])
        return rows.sum(axis=0), rows.shape[0]
    finally:
        _remove_stub_pyspark()


def test_spark_to_local_shards_across_ranks():
    import pickle

    rng = np.random.default_rng(0)
    X = rng.normal(size=(320, 4))
    payload = pickle.dumps(X)
    results = run_distributed(_dist_spark_fit, world_size=2, args=(payload,))
    total_rows = sum(r[1] for r in results)
    assert total_rows == 320
    # both ranks hold NON-overlapping, non-empty shards covering all rows
    assert all(r[1] > 0 for r in results)
    np.testing.assert_allclose(sum(r[0] for r in results), X.sum(axis=0), atol=1e-9)


def test_collect_as_arrow_fallback(stub_pyspark):
    """Spark 3.x has no toArrow(); the bridge must use _collect_as_arrow."""
    import pyarrow as pa

    from spark_rapids_ml_amd.spark.bridge import spark_to_local

    table, X = _blob_table()

    class Spark3DF(stub_pyspark):
        def __init__(self, t):
            super().__init__(t)
            self.used_collect_as_arrow = False

        toArrow = property()  # hasattr(..., "toArrow") -> False via raising

        def _collect_as_arrow(self):
            self.used_collect_as_arrow = True
            return self._table.to_batches()

    # property() without fget raises AttributeError on access -> hasattr False
    sdf = Spark3DF(table)
    assert not hasattr(sdf, "toArrow")
    local = spark_to_local(sdf, shard=False)
    assert sdf.used_collect_as_arrow
    np.testing.assert_allclose(np.asarray(local["features"]), X)


def test_no_code_change_flow_proxy_plus_bridge(stub_pyspark):
    """The complete reference user story: accelerate_pyspark() proxies the
    import, the class fits a (stub) pyspark DataFrame through the Arrow
    bridge, and the model saves in stock Spark format."""
    import importlib

    from spark_rapids_ml_amd.install import accelerate_pyspark

    accelerate_pyspark()
    clustering = importlib.import_module("pyspark.ml.clustering")

    table, X = _blob_table()
    sdf = stub_pyspark(table)
    model = clustering.KMeans(k=3, maxIter=10, seed=1).fit(sdf)
    assert np.asarray(model.cluster_centers_).shape == (3, 6)

    import tempfile

    with tempfile.TemporaryDirectory() as td:
        p = f"{td}/km"
        model.saveAsSparkModel(p)
        from spark_rapids_ml_amd.spark import spark_model_class

        assert spark_model_class(p) == "org.apache.spark.ml.clustering.KMeansModel"

    out = model.transform(sdf)
    assert type(out).__module__.startswith("pyspark.")
