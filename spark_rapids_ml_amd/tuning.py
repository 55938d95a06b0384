"""CrossValidator: k-fold CV with single-pass multi-model fit
(reference tuning.py, 186 LoC).

The reference fits ALL param maps in one data pass per fold via fitMultiple
and evaluates them in one transform job (reference tuning.py:123-130);
folds run on a driver thread pool. Here the SPMD runtime makes thread-level
fold parallelism a collectives hazard (concurrent all-reduces would
interleave), so folds run sequentially — the single-pass property per fold
is preserved, which is where the reference's speedup comes from.
"""

from __future__ import annotations

import numpy as np
from typing import Any, Dict, List, Optional, Tuple

from .core import Estimator, Model
from .data import DataFrame
from .evaluation import Evaluator
from .params import Param, Params, TypeConverters
from .parallel.context import get_comm
from .utils import get_logger


class ParamGridBuilder:
    """pyspark.ml.tuning.ParamGridBuilder equivalent."""

    def __init__(self) -> None:
        self._grid: Dict[Any, List[Any]] = {}

    def addGrid(self, param: Any, values: List[Any]) -> "ParamGridBuilder":
        self._grid[param] = list(values)
        return self

    def baseOn(self, *args: Tuple[Any, Any]) -> "ParamGridBuilder":
        for p, v in args:
            self._grid[p] = [v]
        return self

    def build(self) -> List[Dict[Any, Any]]:
        import itertools

        keys = list(self._grid.keys())
        maps = []
        for combo in itertools.product(*[self._grid[k] for k in keys]):
            maps.append(dict(zip(keys, combo)))
        return maps or [{}]


class CrossValidator(Params):
    """k-fold cross validation (reference CrossValidator, tuning.py:56+)."""

    numFolds = Param("cv", "numFolds", "number of folds.", TypeConverters.toInt)
    seed = Param("cv", "seed", "fold-split seed.", TypeConverters.toInt)
    parallelism = Param("cv", "parallelism", "accepted for API parity; folds run sequentially (SPMD collectives).", TypeConverters.toInt)
    collectSubModels = Param("cv", "collectSubModels", "keep all sub-models.", TypeConverters.toBoolean)
    foldCol = Param("cv", "foldCol", "user-supplied integer fold column (pyspark CV parity).", TypeConverters.toString)

    def __init__(
        self,
        estimator: Optional[Estimator] = None,
        estimatorParamMaps: Optional[List[Dict[Any, Any]]] = None,
        evaluator: Optional[Evaluator] = None,
        numFolds: int = 3,
        seed: int = 42,
        parallelism: int = 1,
        collectSubModels: bool = False,
        foldCol: str = "",
    ) -> None:
        super().__init__()
        self._setDefault(numFolds=3, seed=42, parallelism=1, collectSubModels=False, foldCol="")
        self._set(
            numFolds=numFolds, seed=seed, parallelism=parallelism,
            collectSubModels=collectSubModels, foldCol=foldCol,
        )
        self._estimator = estimator
        self._est_param_maps = estimatorParamMaps or [{}]
        self._evaluator = evaluator

    def setEstimator(self, est: Estimator) -> "CrossValidator":
        self._estimator = est
        return self

    def setEstimatorParamMaps(self, maps: List[Dict[Any, Any]]) -> "CrossValidator":
        self._est_param_maps = maps
        return self

    def setEvaluator(self, ev: Evaluator) -> "CrossValidator":
        self._evaluator = ev
        return self

    def setNumFolds(self, value: int) -> "CrossValidator":
        return self._set(numFolds=value)  # type: ignore[return-value]

    def setCollectSubModels(self, value: bool) -> "CrossValidator":
        return self._set(collectSubModels=value)  # type: ignore[return-value]

    def getEstimator(self) -> Estimator:
        return self._estimator

    def getEvaluator(self) -> Evaluator:
        return self._evaluator

    def getEstimatorParamMaps(self) -> List[Dict[Any, Any]]:
        return self._est_param_maps

    def save(self, path: str) -> None:
        """Persist estimator, evaluator, grid and CV params (the reference
        relies on pyspark's CV persistence + a load-time classpath fix,
        tuning.py:159-186)."""
        import json
        import os

        comm = get_comm()
        self._estimator.save(os.path.join(path, "estimator"))
        if comm.rank == 0:
            meta = {
                "estimator_class": f"{type(self._estimator).__module__}."
                f"{type(self._estimator).__qualname__}",
                "evaluator_class": f"{type(self._evaluator).__module__}."
                f"{type(self._evaluator).__qualname__}",
                "evaluator_params": {
                    p.name: self._evaluator.getOrDefault(p.name)
                    for p in self._evaluator.params
                    if self._evaluator.isDefined(p.name)
                },
                "param_maps": [
                    {(p.name if isinstance(p, Param) else str(p)): v for p, v in pm.items()}
                    for pm in self._est_param_maps
                ],
                "numFolds": self.getOrDefault("numFolds"),
                "seed": self.getOrDefault("seed"),
                "collectSubModels": self.getOrDefault("collectSubModels"),
            }
            with open(os.path.join(path, "cv_estimator_metadata.json"), "w") as f:
                json.dump(meta, f)
        comm.barrier()

    @classmethod
    def load(cls, path: str) -> "CrossValidator":
        import importlib
        import json
        import os

        with open(os.path.join(path, "cv_estimator_metadata.json")) as f:
            meta = json.load(f)

        def _resolve(classpath: str):
            mod, name = classpath.rsplit(".", 1)
            return getattr(importlib.import_module(mod), name)

        est = _resolve(meta["estimator_class"]).load(os.path.join(path, "estimator"))
        ev = _resolve(meta["evaluator_class"])(**meta["evaluator_params"])
        maps = [
            {est.getParam(name): v for name, v in pm.items()}
            for pm in meta["param_maps"]
        ]
        return cls(
            estimator=est,
            estimatorParamMaps=maps,
            evaluator=ev,
            numFolds=meta["numFolds"],
            seed=meta["seed"],
            collectSubModels=meta.get("collectSubModels", False),
        )

    def _kfold(self, df: DataFrame) -> List[Tuple[DataFrame, DataFrame]]:
        """Deterministic per-row fold assignment on each rank's local shard
        (Spark assigns by rand(seed) — same semantics, shard-local)."""
        comm = get_comm()
        n_folds = self.getOrDefault("numFolds")
        fold_col = self.getOrDefault("foldCol")
        if fold_col:
            assign = np.asarray(df[fold_col]).astype(np.int64)
            if assign.size and (assign.min() < 0 or assign.max() >= n_folds):
                raise ValueError(
                    f"foldCol values must lie in [0, numFolds={n_folds})"
                )
        else:
            seed = self.getOrDefault("seed")
            rng = np.random.default_rng(seed + 1000003 * comm.rank)
            assign = rng.integers(0, n_folds, size=df.num_rows)
        folds = []
        for f in range(n_folds):
            test_idx = np.nonzero(assign == f)[0]
            train_idx = np.nonzero(assign != f)[0]
            folds.append((df.take_local(train_idx), df.take_local(test_idx)))
        return folds

    def fit(self, df: DataFrame) -> "CrossValidatorModel":
        assert self._estimator is not None and self._evaluator is not None
        logger = get_logger(self.__class__)
        epm = self._est_param_maps
        n_folds = self.getOrDefault("numFolds")
        metrics = np.zeros((len(epm), n_folds))
        collect = bool(self.getOrDefault("collectSubModels"))
        sub = [[None] * len(epm) for _ in range(n_folds)] if collect else None
        from .core import Model

        for fold, (train, test) in enumerate(self._kfold(df)):
            # single data pass over all param maps (reference tuning.py:123-130):
            # fitMultiple trains every map against one load, _combine +
            # _transformEvaluate scores them all in ONE feature extraction
            # (reference one-job multi-model evaluate, core.py:1572-1693)
            models = [m for _, m in self._estimator.fitMultiple(train, epm)]
            combined = Model._combine(models)
            fold_metrics = combined._transformEvaluate(test, self._evaluator)
            metrics[:, fold] = fold_metrics
            if collect:
                for idx, model in enumerate(models):
                    sub[fold][idx] = model
        avg = metrics.mean(axis=1)
        std = metrics.std(axis=1)
        best = int(np.argmax(avg) if self._evaluator.isLargerBetter() else np.argmin(avg))
        logger.info(f"CV best param map index {best}: avg metric {avg[best]:.6f}")
        best_model = self._estimator.fit(df, epm[best])
        cvm = CrossValidatorModel(
            bestModel=best_model, avgMetrics=avg.tolist(), stdMetrics=std.tolist(),
            subModels=sub,
        )
        self._copyValues(cvm)
        return cvm


class CrossValidatorModel(Params):
    """Holds the best model + per-param-map metrics (reference
    tuning.py:141-157). Persistence stores the best model under
    bestModel/ plus metrics JSON (reference load() resolves the concrete
    model class the same way, tuning.py:159-186)."""

    def __init__(
        self,
        bestModel: Optional[Model] = None,
        avgMetrics: Optional[List[float]] = None,
        stdMetrics: Optional[List[float]] = None,
        subModels: Optional[List[List[Model]]] = None,
    ) -> None:
        super().__init__()
        self.bestModel = bestModel
        self.subModels = subModels
        self.avgMetrics = avgMetrics or []
        self.stdMetrics = stdMetrics or []

    def transform(self, df: DataFrame) -> DataFrame:
        return self.bestModel.transform(df)

    def save(self, path: str) -> None:
        import json
        import os

        comm = get_comm()
        self.bestModel.write().overwrite().save(os.path.join(path, "bestModel"))
        if comm.rank == 0:
            meta = {
                "class": f"{type(self.bestModel).__module__}."
                f"{type(self.bestModel).__qualname__}",
                "avgMetrics": self.avgMetrics,
                "stdMetrics": self.stdMetrics,
            }
            with open(os.path.join(path, "cv_metadata.json"), "w") as f:
                json.dump(meta, f)
        comm.barrier()

    @classmethod
    def load(cls, path: str) -> "CrossValidatorModel":
        import importlib
        import json
        import os

        with open(os.path.join(path, "cv_metadata.json")) as f:
            meta = json.load(f)
        mod_name, cls_name = meta["class"].rsplit(".", 1)
        model_cls = getattr(importlib.import_module(mod_name), cls_name)
        best = model_cls.load(os.path.join(path, "bestModel"))
        return cls(
            bestModel=best,
            avgMetrics=meta["avgMetrics"],
            stdMetrics=meta["stdMetrics"],
        )
