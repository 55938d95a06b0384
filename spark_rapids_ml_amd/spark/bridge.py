"""pyspark.sql.DataFrame bridge: fit/transform on real Spark DataFrames.

The reference's defining entry path is a pyspark DataFrame flowing into
`Estimator.fit` (reference core.py:742-1013: barrier `mapInPandas`, one task
per GPU). This engine is SPMD (one process per GPU, torchrun), so the bridge
is a **driver-side Arrow export**: every rank evaluates the same program,
pulls the Spark DataFrame as Arrow batches, and keeps its round-robin share
of the batches as its local shard. Model.transform on a pyspark DataFrame
computes locally and rebuilds a pyspark DataFrame via createDataFrame.

Everything is import-gated: with no pyspark installed this module only
provides `is_pyspark_dataframe` (always False) and the conversion helpers
raise. VectorUDT feature columns are unwrapped with
`pyspark.ml.functions.vector_to_array` before Arrow collection (Arrow does
not ship UDTs; the reference unwraps the same way, core.py:220-265).
"""

from __future__ import annotations

from typing import Any, List, Optional, Sequence

import numpy as np


def _pyspark_available() -> bool:
    try:
        import pyspark  # noqa: F401

        return True
    except ImportError:
        return False


def is_pyspark_dataframe(obj: Any) -> bool:
    """True for pyspark.sql.DataFrame (classic or connect), without
    importing pyspark when it isn't installed."""
    mod = type(obj).__module__ or ""
    if not mod.startswith("pyspark."):
        return False
    return type(obj).__name__ == "DataFrame"


def _vector_columns(sdf) -> List[str]:
    """Names of VectorUDT columns in a pyspark DataFrame's schema."""
    out = []
    for field in sdf.schema.fields:
        t = field.dataType
        if type(t).__name__ == "VectorUDT":
            out.append(field.name)
    return out


def _collect_arrow_batches(sdf) -> List[Any]:
    """Collect a pyspark DataFrame as a list of Arrow RecordBatches
    (Spark 4: toArrow(); Spark 3.x: _collect_as_arrow())."""
    if hasattr(sdf, "toArrow"):
        table = sdf.toArrow()
        return table.to_batches()
    if hasattr(sdf, "_collect_as_arrow"):
        return sdf._collect_as_arrow()
    # last resort: pandas hop
    import pyarrow as pa

    pdf = sdf.toPandas()
    return pa.Table.from_pandas(pdf).to_batches()


def spark_to_local(sdf, shard: bool = True):
    """Convert a pyspark DataFrame into this package's columnar DataFrame.

    VectorUDT columns are unwrapped to array<float/double> columns first
    (keeping their names) so they flow through Arrow; with `shard` and
    world_size>1 each rank keeps batches round-robin by index (batch i goes
    to rank i % world) — the Spark-partition analog of the reference's
    repartition(num_workers) + barrier tasks (reference core.py:771-772)."""
    if not _pyspark_available():
        raise RuntimeError("pyspark is not installed; cannot convert a Spark DataFrame")

    from ..data import DataFrame as LocalDF
    from ..parallel.context import get_comm

    vec_cols = _vector_columns(sdf)
    if vec_cols:
        from pyspark.ml.functions import vector_to_array
        from pyspark.sql import functions as F

        proj = [
            vector_to_array(F.col(c)).alias(c) if c in vec_cols else F.col(c)
            for c in sdf.columns
        ]
        sdf = sdf.select(*proj)

    batches = _collect_arrow_batches(sdf)
    comm = get_comm()
    if shard and comm.world_size > 1:
        mine = [b for i, b in enumerate(batches) if i % comm.world_size == comm.rank]
    else:
        mine = batches

    import pyarrow as pa

    if mine:
        table = pa.Table.from_batches(mine)
    elif batches:
        table = pa.Table.from_batches(batches).slice(0, 0)
    else:
        raise ValueError("Spark DataFrame produced no Arrow batches")
    return LocalDF._from_arrow(table, vector_cols=vec_cols)


def local_to_spark(df, spark, schema_hint: Optional[Sequence[str]] = None):
    """Build a pyspark DataFrame from this package's columnar DataFrame on
    the given SparkSession (used by Model.transform to hand results back)."""
    if not _pyspark_available():
        raise RuntimeError("pyspark is not installed")
    import pandas as pd

    cols = {}
    for name in df.columns:
        col = df[name]
        arr = np.asarray(col)
        if arr.ndim == 2:
            cols[name] = list(arr)  # array<double> rows
        else:
            cols[name] = arr
    return spark.createDataFrame(pd.DataFrame(cols))


def fit_on_spark(estimator, sdf):
    """Estimator.fit entry for pyspark DataFrames."""
    local = spark_to_local(sdf, shard=True)
    return estimator._fit(local)


def transform_on_spark(model, sdf):
    """Model.transform entry for pyspark DataFrames: compute locally
    (unsharded — transform is row-local, reference core.py:1797 runs it as
    plain tasks) and return a pyspark DataFrame on the input's session."""
    local = spark_to_local(sdf, shard=False)
    out = model.transform(local)
    spark = getattr(sdf, "sparkSession", None) or sdf.sql_ctx.sparkSession
    return local_to_spark(out, spark)
