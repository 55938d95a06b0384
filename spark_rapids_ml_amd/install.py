"""No-import-change acceleration (reference install.py:22-81).

The reference installs proxy modules over `pyspark.ml.*` so unmodified
pyspark scripts pick up the GPU classes. The analog here: scripts written
against the REFERENCE package (`import spark_rapids_ml` / its submodules
feature, clustering, classification, regression, knn, umap, tuning,
pipeline) run unmodified on this framework — `install_aliases()` registers
`spark_rapids_ml` (and, when pyspark is absent, `pyspark.ml`-shaped alias
modules) in sys.modules pointing at the spark_rapids_ml_amd classes.
"""

from __future__ import annotations

import sys
import types
from typing import Dict, List


_SUBMODULE_EXPORTS: Dict[str, List[str]] = {
    "feature": ["PCA", "PCAModel"],
    "clustering": ["KMeans", "KMeansModel", "DBSCAN", "DBSCANModel"],
    "classification": [
        "LogisticRegression",
        "LogisticRegressionModel",
        "RandomForestClassifier",
        "RandomForestClassificationModel",
    ],
    "regression": [
        "LinearRegression",
        "LinearRegressionModel",
        "RandomForestRegressor",
        "RandomForestRegressionModel",
    ],
    "knn": [
        "NearestNeighbors",
        "NearestNeighborsModel",
        "ApproximateNearestNeighbors",
        "ApproximateNearestNeighborsModel",
    ],
    "umap": ["UMAP", "UMAPModel"],
    "tuning": ["CrossValidator", "CrossValidatorModel"],
    "pipeline": ["Pipeline", "PipelineModel", "NoOpTransformer"],
    "evaluation": [
        "RegressionEvaluator",
        "MulticlassClassificationEvaluator",
        "BinaryClassificationEvaluator",
    ],
    "metrics": [],   # filled from the metrics package below
    "core": [],      # Estimator/Model base aliases
    "params": [],    # Param surface
}


def _make_module(name: str, exports: Dict[str, object]) -> types.ModuleType:
    mod = types.ModuleType(name)
    for k, v in exports.items():
        setattr(mod, k, v)
    mod.__all__ = list(exports)
    return mod


def install_aliases(alias_root: str = "spark_rapids_ml") -> None:
    """Register `spark_rapids_ml` alias modules over this package."""
    import spark_rapids_ml_amd as root

    from .tuning import ParamGridBuilder

    root_exports = {name: getattr(root, name) for name in root.__all__}
    root_mod = _make_module(alias_root, root_exports)
    sys.modules[alias_root] = root_mod

    for sub, names in _SUBMODULE_EXPORTS.items():
        exports = {}
        for n in names:
            if hasattr(root, n):
                exports[n] = getattr(root, n)
        if sub == "tuning":
            exports["ParamGridBuilder"] = ParamGridBuilder
        if sub == "pipeline":
            from .pipeline import VectorAssembler

            exports["VectorAssembler"] = VectorAssembler
        if sub == "evaluation":
            from . import evaluation

            for n in names:
                exports[n] = getattr(evaluation, n)
        if sub == "metrics":
            from .metrics import MulticlassMetrics, RegressionMetrics

            exports["MulticlassMetrics"] = MulticlassMetrics
            exports["RegressionMetrics"] = RegressionMetrics
        if sub == "core":
            from .core import Estimator, Model

            exports["Estimator"] = exports["_CumlEstimator"] = Estimator
            exports["Model"] = exports["_CumlModel"] = Model
        if sub == "params":
            from .params import Param, Params, TypeConverters, DictTypeConverters

            exports.update(
                Param=Param, Params=Params, TypeConverters=TypeConverters,
                DictTypeConverters=DictTypeConverters,
            )
        m = _make_module(f"{alias_root}.{sub}", exports)
        sys.modules[f"{alias_root}.{sub}"] = m
        setattr(root_mod, sub, m)


class _PysparkProxyModule(types.ModuleType):
    """Module proxy over a real pyspark.ml submodule: accelerated class
    lookups resolve to the spark_rapids_ml_amd classes, everything else
    falls through to stock pyspark (reference install.py:22-81)."""

    def __init__(self, name: str, orig: types.ModuleType, accel: Dict[str, object]):
        super().__init__(name, getattr(orig, "__doc__", None))
        self._srml_orig = orig
        self._srml_accel = accel
        self.__file__ = getattr(orig, "__file__", None)
        self.__package__ = getattr(orig, "__package__", None)

    def __getattr__(self, name: str):
        accel = self.__dict__["_srml_accel"]
        if name in accel:
            return accel[name]
        return getattr(self.__dict__["_srml_orig"], name)


# pyspark.ml submodules the reference proxies (install.py:22-49) and the
# accelerated class names each one redirects
_PYSPARK_PROXY: Dict[str, List[str]] = {
    "feature": ["PCA", "PCAModel"],
    "clustering": ["KMeans", "KMeansModel"],
    "classification": [
        "LogisticRegression",
        "LogisticRegressionModel",
        "RandomForestClassifier",
        "RandomForestClassificationModel",
    ],
    "regression": [
        "LinearRegression",
        "LinearRegressionModel",
        "RandomForestRegressor",
        "RandomForestRegressionModel",
    ],
    "tuning": ["CrossValidator", "CrossValidatorModel"],
}


def accelerate_pyspark() -> List[str]:
    """Install proxy modules over the REAL `pyspark.ml.{feature,clustering,
    classification,regression,tuning}` so unmodified pyspark scripts resolve
    the accelerated classes (reference install.py:22-81). Returns the list
    of proxied module names. Raises ImportError when pyspark is absent."""
    import importlib

    import pyspark.ml as pml  # raises ImportError without pyspark

    import spark_rapids_ml_amd as root

    proxied: List[str] = []
    for sub, names in _PYSPARK_PROXY.items():
        full = f"pyspark.ml.{sub}"
        orig = sys.modules.get(full) or importlib.import_module(full)
        if isinstance(orig, _PysparkProxyModule):
            proxied.append(full)
            continue
        accel = {n: getattr(root, n) for n in names if hasattr(root, n)}
        if sub == "tuning":
            from .tuning import CrossValidator, CrossValidatorModel

            accel["CrossValidator"] = CrossValidator
            accel["CrossValidatorModel"] = CrossValidatorModel
        proxy = _PysparkProxyModule(full, orig, accel)
        sys.modules[full] = proxy
        setattr(pml, sub, proxy)
        proxied.append(full)
    return proxied


def main() -> None:  # console-script parity with reference install.py
    install_aliases()
    try:
        proxied = accelerate_pyspark()
        print(f"pyspark.ml proxies installed: {', '.join(proxied)}")
    except ImportError:
        print("pyspark not installed; only spark_rapids_ml aliases active")
    print("spark_rapids_ml -> spark_rapids_ml_amd aliases installed")


if __name__ == "__main__":
    main()
