"""Pipeline with the VectorAssembler bypass (reference pipeline.py, 159 LoC).

The reference replaces a VectorAssembler directly feeding a GPU estimator
with a NoOpTransformer and passes the scalar columns as featuresCols instead
(reference pipeline.py:85-119) — skipping the array materialization. The
same rewrite applies here: assembly into a 2-D features column is pure
copy-bandwidth, and the estimators ingest multi-column input natively.
"""

from __future__ import annotations

from typing import Any, List, Optional

import numpy as np

from .core import Estimator, Model
from .data import DataFrame
from .params import HasOutputCol, Param, Params, TypeConverters


class Transformer(Params):
    def transform(self, df: DataFrame) -> DataFrame:
        raise NotImplementedError


class VectorAssembler(Transformer, HasOutputCol):
    """Assemble scalar columns into one 2-D features column
    (pyspark.ml.feature.VectorAssembler equivalent)."""

    inputCols = Param("va", "inputCols", "input columns.", TypeConverters.toListString)

    def __init__(self, inputCols: Optional[List[str]] = None, outputCol: str = "features"):
        super().__init__()
        if inputCols is not None:
            self._set(inputCols=inputCols)
        self._set(outputCol=outputCol)

    def setInputCols(self, value: List[str]) -> "VectorAssembler":
        return self._set(inputCols=value)  # type: ignore[return-value]

    def setOutputCol(self, value: str) -> "VectorAssembler":
        return self._set(outputCol=value)  # type: ignore[return-value]

    def getInputCols(self) -> List[str]:
        return self.getOrDefault("inputCols")

    def transform(self, df: DataFrame) -> DataFrame:
        cols = [np.asarray(df[c]).reshape(len(df), -1) for c in self.getInputCols()]
        feat = np.column_stack(cols).astype(np.float32) if len(df) else np.zeros((0, len(cols)), np.float32)
        return df.with_column(self.getOrDefault("outputCol"), feat)


class NoOpTransformer(Transformer):
    """Stand-in for a bypassed VectorAssembler (reference pipeline.py:42-56)."""

    def transform(self, df: DataFrame) -> DataFrame:
        return df


def _save_stages(path: str, stages: List[Any]) -> None:
    import json
    import os

    from .parallel.context import get_comm

    comm = get_comm()
    meta: List[Any] = []
    for i, st in enumerate(stages):
        cls_name = f"{type(st).__module__}.{type(st).__qualname__}"
        if isinstance(st, (Estimator, Model)):
            st.save(os.path.join(path, f"stage_{i}"))
            meta.append({"class": cls_name, "kind": "dir"})
        else:
            params = {
                p.name: st.getOrDefault(p.name)
                for p in st.params
                if st.isSet(p.name) or st.hasDefault(p.name)
            }
            meta.append({"class": cls_name, "kind": "params", "params": params})
    if comm.rank == 0:
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "pipeline_metadata.json"), "w") as f:
            json.dump(meta, f)
    comm.barrier()


def _load_stages(path: str) -> List[Any]:
    import importlib
    import json
    import os

    with open(os.path.join(path, "pipeline_metadata.json")) as f:
        meta = json.load(f)
    stages: List[Any] = []
    for i, ent in enumerate(meta):
        mod_name, cls_name = ent["class"].rsplit(".", 1)
        st_cls = getattr(importlib.import_module(mod_name), cls_name)
        if ent["kind"] == "dir":
            stages.append(st_cls.load(os.path.join(path, f"stage_{i}")))
        else:
            st = st_cls()
            for k, v in ent["params"].items():
                st._set(**{k: v})
            stages.append(st)
    return stages


class Pipeline(Params):
    """Sequential stages with the assembler bypass (reference Pipeline,
    pipeline.py:59+). save/load mirror pyspark Pipeline persistence:
    estimator/model stages persist via their own writers, plain
    transformers via a params JSON."""

    def __init__(self, stages: Optional[List[Any]] = None) -> None:
        super().__init__()
        self._stages = stages or []

    def setStages(self, stages: List[Any]) -> "Pipeline":
        self._stages = stages
        return self

    def getStages(self) -> List[Any]:
        return self._stages

    def _rewrite_stages(self) -> List[Any]:
        """Replace VectorAssembler -> GPU-estimator pairs with NoOp +
        featuresCols wiring (reference pipeline.py:85-119)."""
        stages = list(self._stages)
        for i in range(len(stages) - 1):
            st, nxt = stages[i], stages[i + 1]
            if (
                isinstance(st, VectorAssembler)
                and isinstance(nxt, Estimator)
                and nxt.hasParam("featuresCols")
                and (
                    not nxt.isSet("featuresCol")
                    or nxt.getOrDefault("featuresCol") == st.getOrDefault("outputCol")
                )
            ):
                nxt._set_params(featuresCols=st.getInputCols())
                stages[i] = NoOpTransformer()
        return stages

    def fit(self, df: DataFrame) -> "PipelineModel":
        stages = self._rewrite_stages()
        fitted: List[Any] = []
        cur = df
        for i, st in enumerate(stages):
            if isinstance(st, Estimator):
                model = st.fit(cur)
                fitted.append(model)
                if i < len(stages) - 1:
                    cur = model.transform(cur)
            elif isinstance(st, (Transformer, Model)):
                fitted.append(st)
                if i < len(stages) - 1:
                    cur = st.transform(cur)
            else:
                raise TypeError(f"stage {st!r} is not an Estimator/Transformer")
        return PipelineModel(fitted)

    def save(self, path: str) -> None:
        _save_stages(path, self._stages)

    @classmethod
    def load(cls, path: str) -> "Pipeline":
        return cls(_load_stages(path))


class PipelineModel(Params):
    def __init__(self, stages: List[Any]) -> None:
        super().__init__()
        self.stages = stages

    def transform(self, df: DataFrame) -> DataFrame:
        cur = df
        for st in self.stages:
            cur = st.transform(cur)
        return cur

    def save(self, path: str) -> None:
        _save_stages(path, self.stages)

    @classmethod
    def load(cls, path: str) -> "PipelineModel":
        return cls(_load_stages(path))
