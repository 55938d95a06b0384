"""Remote fit/transform service — the Spark Connect plugin equivalent.

The reference ships a JVM MLBackendPlugin + a Python worker
(`connect_plugin.py:96-245`) so Spark Connect clients get no-code-change GPU
acceleration: the server receives an operator name + params + a data
reference, runs fit (returning model attributes as JSON) or transform
(returning a result reference). The MI355X-native analog is an HTTP service
(FastAPI) with the same three operations over parquet data references:

  POST /fit        {estimator, params, data_path}        -> {model_id, attributes}
  POST /transform  {model_id, data_path, output_path}    -> {output_path, columns}
  GET  /models                                            -> registry
  POST /models/{id}/save {path}                           -> persisted model

Run: python -m spark_rapids_ml_amd.connect_server --port 8571
"""

import argparse
import importlib
import uuid
from typing import Any, Dict, Optional

from .data import DataFrame
from .core import Estimator, Model

_ESTIMATORS = {
    "KMeans": "spark_rapids_ml_amd.models.clustering",
    "DBSCAN": "spark_rapids_ml_amd.models.clustering",
    "PCA": "spark_rapids_ml_amd.models.feature",
    "LinearRegression": "spark_rapids_ml_amd.models.regression",
    "RandomForestRegressor": "spark_rapids_ml_amd.models.regression",
    "LogisticRegression": "spark_rapids_ml_amd.models.classification",
    "RandomForestClassifier": "spark_rapids_ml_amd.models.classification",
    "NearestNeighbors": "spark_rapids_ml_amd.models.knn",
    "ApproximateNearestNeighbors": "spark_rapids_ml_amd.models.knn",
    "UMAP": "spark_rapids_ml_amd.models.umap",
}


class ModelRegistry:
    """In-memory model registry (the reference keeps a py4j object registry,
    Utils.scala:57-105)."""

    def __init__(self) -> None:
        self._models: Dict[str, Model] = {}

    def put(self, model: Model) -> str:
        mid = uuid.uuid4().hex[:12]
        self._models[mid] = model
        return mid

    def get(self, mid: str) -> Model:
        if mid not in self._models:
            raise KeyError(f"unknown model id {mid}")
        return self._models[mid]

    def list(self) -> Dict[str, str]:
        return {mid: type(m).__name__ for mid, m in self._models.items()}


def _check_path(path: str) -> str:
    """Filesystem confinement: data/output paths must resolve under
    SRML_SERVER_DATA_ROOT (defaults to the server's CWD, so a reachable
    client cannot read/write arbitrary paths as the service user; set the
    env to widen or relocate the root)."""
    import os

    root = os.environ.get("SRML_SERVER_DATA_ROOT") or os.getcwd()
    real = os.path.realpath(path)
    root_real = os.path.realpath(root)
    if real != root_real and not real.startswith(root_real + os.sep):
        raise ValueError(f"path {path!r} outside SRML_SERVER_DATA_ROOT ({root!r})")
    return path


def _resolve_estimator(name: str, params: Dict[str, Any]) -> Estimator:
    if name not in _ESTIMATORS:
        raise ValueError(f"unsupported estimator {name!r}")
    mod = importlib.import_module(_ESTIMATORS[name])
    cls = getattr(mod, name)
    return cls(**params)


def _attrs_json(model: Model) -> Dict[str, Any]:
    """Model attributes with arrays summarized (full arrays persist via
    /models/{id}/save; the reference returns attribute JSON the same way,
    connect_plugin.py:139)."""
    import numpy as np

    out: Dict[str, Any] = {}
    for k, v in model._get_model_attributes().items():
        if isinstance(v, np.ndarray):
            if v.size <= 10000:
                out[k] = v.tolist()
            else:
                out[k] = {"__array__": True, "shape": list(v.shape), "dtype": str(v.dtype)}
        else:
            out[k] = v
    return out


try:  # pydantic request schemas at module scope so FastAPI resolves them
    from pydantic import BaseModel as _PydModel

    class FitRequest(_PydModel):
        estimator: str
        params: Dict[str, Any] = {}
        data_path: str

    class TransformRequest(_PydModel):
        model_id: str
        data_path: str
        output_path: str

    class SaveRequest(_PydModel):
        path: str

except ImportError:  # pragma: no cover - server extras missing
    FitRequest = TransformRequest = SaveRequest = None  # type: ignore


def create_app(registry: Optional[ModelRegistry] = None):
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="spark_rapids_ml_amd connect server")
    reg = registry or ModelRegistry()
    app.state.registry = reg

    @app.post("/fit")
    def fit(req: FitRequest):
        try:
            est = _resolve_estimator(req.estimator, req.params)
            df = DataFrame.read_parquet(_check_path(req.data_path), vector_cols=["features"])
            model = est.fit(df)
            mid = reg.put(model)
            return {"model_id": mid, "attributes": _attrs_json(model)}
        except (ValueError, KeyError, FileNotFoundError) as e:
            raise HTTPException(status_code=400, detail=str(e))

    @app.post("/transform")
    def transform(req: TransformRequest):
        try:
            model = reg.get(req.model_id)
            df = DataFrame.read_parquet(_check_path(req.data_path), vector_cols=["features"])
            out = model.transform(df)
            out.write_parquet(_check_path(req.output_path))
            return {"output_path": req.output_path, "columns": out.columns}
        except (ValueError, KeyError, FileNotFoundError) as e:
            raise HTTPException(status_code=400, detail=str(e))

    @app.get("/models")
    def models():
        return reg.list()

    @app.post("/models/{model_id}/save")
    def save(model_id: str, req: SaveRequest):
        try:
            reg.get(model_id).write().overwrite().save(_check_path(req.path))
            return {"path": req.path}
        except KeyError as e:
            raise HTTPException(status_code=404, detail=str(e))

    return app


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8571)
    args = ap.parse_args()
    import uvicorn

    uvicorn.run(create_app(), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
