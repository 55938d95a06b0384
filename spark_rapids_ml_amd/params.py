"""Param system: a self-contained pyspark.ml.param-compatible implementation
plus the Spark-param <-> native-param two-way sync machinery.

The reference delegates to pyspark.ml.param and layers `_CumlClass`/`_CumlParams`
on top (reference params.py:169-237, 430-487). pyspark is not a dependency
here, so the Param/Params surface itself is reimplemented with the same
semantics (Param identity, default vs user-set maps, getOrDefault, copy) and
the sync layer follows the reference's behavior:

- `_param_mapping()` maps a Spark param name to a native param name, with two
  sentinels: map to ``None`` -> unsupported, raise on a non-default value; map
  to ``""`` -> silently ignored (reference params.py:169-201).
- `_param_value_mapping()` maps Spark param *values* to native values through
  per-param lambdas; a lambda returning ``None`` means that value is
  unsupported (reference params.py:203-237).
- `_set_params` writes through to both maps (reference params.py:430-487).
"""

from __future__ import annotations

import copy as _copy
from typing import Any, Callable, Dict, List, Optional, TypeVar, Union

P = TypeVar("P", bound="Params")


class TypeConverters:
    """Value coercion helpers mirroring pyspark.ml.param.TypeConverters."""

    @staticmethod
    def identity(value: Any) -> Any:
        return value

    @staticmethod
    def toInt(value: Any) -> int:
        if isinstance(value, bool):
            raise TypeError(f"Could not convert {value} to int")
        return int(value)

    @staticmethod
    def toFloat(value: Any) -> float:
        return float(value)

    @staticmethod
    def toBoolean(value: Any) -> bool:
        if isinstance(value, bool):
            return value
        raise TypeError(f"Boolean Param requires value of type bool. Found {type(value)}.")

    @staticmethod
    def toString(value: Any) -> str:
        return str(value)

    @staticmethod
    def toList(value: Any) -> list:
        if isinstance(value, (list, tuple)):
            return list(value)
        import numpy as np

        if isinstance(value, np.ndarray):
            return value.tolist()
        raise TypeError(f"Could not convert {value} to list")

    @staticmethod
    def toListFloat(value: Any) -> List[float]:
        return [float(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toListInt(value: Any) -> List[int]:
        return [int(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toListString(value: Any) -> List[str]:
        return [str(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toDict(value: Any) -> dict:
        if isinstance(value, dict):
            return value
        raise TypeError(f"Could not convert {value} to dict")


class DictTypeConverters(TypeConverters):
    """Extra converters used by the reference for dict-typed params
    (reference params.py DictTypeConverters)."""

    @staticmethod
    def _toDict(value: Any) -> dict:
        return TypeConverters.toDict(value)


class Param:
    """A parameter with self-contained documentation (pyspark.ml.param.Param)."""

    def __init__(
        self,
        parent: Union["Params", str],
        name: str,
        doc: str,
        typeConverter: Optional[Callable[[Any], Any]] = None,
    ):
        self.parent = parent.uid if isinstance(parent, Params) else str(parent)
        self.name = name
        self.doc = doc
        self.typeConverter = typeConverter or TypeConverters.identity

    def _copy_new_parent(self, parent: "Params") -> "Param":
        p = _copy.copy(self)
        p.parent = parent.uid
        return p

    def __str__(self) -> str:
        return f"{self.parent}__{self.name}"

    def __repr__(self) -> str:
        return f"Param(parent={self.parent!r}, name={self.name!r}, doc={self.doc!r})"

    def __hash__(self) -> int:
        return hash(str(self))

    def __eq__(self, other: Any) -> bool:
        return isinstance(other, Param) and self.parent == other.parent and self.name == other.name


_uid_counters: Dict[str, int] = {}


def _gen_uid(cls_name: str) -> str:
    n = _uid_counters.get(cls_name, 0)
    _uid_counters[cls_name] = n + 1
    import random

    return f"{cls_name}_{random.getrandbits(48):012x}"


class Params:
    """Base class holding params, mirroring pyspark.ml.param.Params semantics:
    a default map set by the component and a user map set via setters; a
    `Param` is identified by (parent uid, name)."""

    def __init__(self, **kwargs: Any) -> None:
        self._paramMap: Dict[Param, Any] = {}
        self._defaultParamMap: Dict[Param, Any] = {}
        self.uid = _gen_uid(self.__class__.__name__)
        self._params: Optional[List[Param]] = None
        self._copy_params()

    @classmethod
    def _declared_params(cls) -> Dict[str, Param]:
        """All Param declarations in the MRO, including ones shadowed by a
        property on a more-derived class (e.g. a model exposing a same-named
        attribute property)."""
        found: Dict[str, Param] = {}
        for klass in reversed(cls.__mro__):
            for name, attr in vars(klass).items():
                if isinstance(attr, Param):
                    found[name] = attr
        return found

    def _copy_params(self) -> None:
        """Bind class-level Param declarations to this instance."""
        self._bound_params: Dict[str, Param] = {}
        for name, attr in self._declared_params().items():
            bound = attr._copy_new_parent(self)
            self._bound_params[name] = bound
            try:
                setattr(self, name, bound)
            except AttributeError:
                pass  # shadowed by a read-only property; getParam still works

    @property
    def params(self) -> List[Param]:
        if self._params is None:
            self._params = sorted(self._bound_params.values(), key=lambda p: p.name)
        return self._params

    def hasParam(self, paramName: str) -> bool:
        return any(p.name == paramName for p in self.params)

    def getParam(self, paramName: str) -> Param:
        for p in self.params:
            if p.name == paramName:
                return p
        raise ValueError(f"{self.__class__.__name__} has no param {paramName!r}")

    def _resolveParam(self, param: Union[str, Param]) -> Param:
        if isinstance(param, str):
            return self.getParam(param)
        if param.parent != self.uid:
            raise ValueError(f"Param {param} does not belong to {self.uid}")
        return param

    def isSet(self, param: Union[str, Param]) -> bool:
        return self._resolveParam(param) in self._paramMap

    def hasDefault(self, param: Union[str, Param]) -> bool:
        return self._resolveParam(param) in self._defaultParamMap

    def isDefined(self, param: Union[str, Param]) -> bool:
        return self.isSet(param) or self.hasDefault(param)

    def get(self, param: Union[str, Param]) -> Any:
        return self.getOrDefault(param)

    def getOrDefault(self, param: Union[str, Param]) -> Any:
        param = self._resolveParam(param)
        if param in self._paramMap:
            return self._paramMap[param]
        return self._defaultParamMap[param]

    def __getattr__(self, name: str) -> Any:
        """Auto-generate pyspark-style `getX()` getters and `setX(v)` setters
        for declared params (pyspark writes these out by hand in shared.py;
        here any param `seed` answers `getSeed()`/`setSeed(v)`, and a
        snake_case param `min_samples` answers `getMinSamples()`). Called only
        when normal attribute lookup fails, so explicit getters/setters/
        properties — including ones that must raise — win."""
        if (name.startswith("get") or name.startswith("set")) and len(name) > 3:
            try:
                bound = object.__getattribute__(self, "_bound_params")
            except AttributeError:
                bound = {}
            camel = name[3].lower() + name[4:]
            import re as _re

            snake = _re.sub(r"(?<!^)([A-Z])", r"_\1", name[3:]).lower()
            target = next((c for c in (camel, name[3:], snake) if c in bound), None)
            if target is not None:
                if name.startswith("get"):
                    return lambda: self.getOrDefault(target)
                setter = getattr(self, "_set_params", None) or self._set
                return lambda value: setter(**{target: value})
        raise AttributeError(
            f"{type(self).__name__!r} object has no attribute {name!r}"
        )

    def _set(self, **kwargs: Any) -> "Params":
        for k, v in kwargs.items():
            p = self.getParam(k)
            if v is not None:
                try:
                    v = p.typeConverter(v)
                except (TypeError, ValueError) as e:
                    raise TypeError(f'Invalid param value given for param "{k}". {e}')
            self._paramMap[p] = v
        return self

    def set(self, param: Union[str, Param], value: Any) -> "Params":
        p = self._resolveParam(param)
        return self._set(**{p.name: value})

    def _setDefault(self, **kwargs: Any) -> "Params":
        for k, v in kwargs.items():
            p = self.getParam(k)
            self._defaultParamMap[p] = v
        return self

    def clear(self, param: Union[str, Param]) -> None:
        p = self._resolveParam(param)
        self._paramMap.pop(p, None)

    def extractParamMap(self, extra: Optional[Dict[Param, Any]] = None) -> Dict[Param, Any]:
        pm = dict(self._defaultParamMap)
        pm.update(self._paramMap)
        if extra:
            pm.update(extra)
        return pm

    def explainParam(self, param: Union[str, Param]) -> str:
        p = self._resolveParam(param)
        values = []
        if self.hasDefault(p):
            values.append(f"default: {self._defaultParamMap[p]}")
        if self.isSet(p):
            values.append(f"current: {self._paramMap[p]}")
        return f"{p.name}: {p.doc} ({', '.join(values) if values else 'undefined'})"

    def explainParams(self) -> str:
        return "\n".join(self.explainParam(p) for p in self.params)

    def copy(self: P, extra: Optional[Dict[Param, Any]] = None) -> P:
        that = _copy.copy(self)
        that._paramMap = dict(self._paramMap)
        that._defaultParamMap = dict(self._defaultParamMap)
        if hasattr(self, "_native_params"):
            that._native_params = dict(self._native_params)  # type: ignore[attr-defined]
        if extra:
            for p, v in extra.items():
                that.set(that.getParam(p.name if isinstance(p, Param) else p), v)
        return that

    def _copyValues(self, to: "Params", extra: Optional[Dict[Param, Any]] = None) -> "Params":
        pm = dict(self._paramMap)
        if extra:
            pm.update(extra)
        for p, v in self._defaultParamMap.items():
            if to.hasParam(p.name):
                to._defaultParamMap[to.getParam(p.name)] = v
        for p, v in pm.items():
            if to.hasParam(p.name):
                to._paramMap[to.getParam(p.name)] = v
        return to


# ---------------------------------------------------------------------------
# Common shared param mixins (subset of pyspark.ml.param.shared used by the
# reference algorithms).
# ---------------------------------------------------------------------------


class HasFeaturesCol(Params):
    featuresCol = Param(
        "shared", "featuresCol", "features column name.", TypeConverters.toString
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(featuresCol="features")

    def getFeaturesCol(self) -> str:
        return self.getOrDefault(self.featuresCol)


class HasFeaturesCols(Params):
    """Param for a list of scalar feature columns, an extension the reference
    adds over Spark (reference params.py:69-93)."""

    featuresCols = Param(
        "shared",
        "featuresCols",
        "features column names for multi-column input.",
        TypeConverters.toListString,
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)

    def getFeaturesCols(self) -> List[str]:
        return self.getOrDefault(self.featuresCols)

    def setFeaturesCols(self, value: List[str]) -> "HasFeaturesCols":
        return self._set(featuresCols=value)  # type: ignore[return-value]


class HasLabelCol(Params):
    labelCol = Param("shared", "labelCol", "label column name.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(labelCol="label")

    def getLabelCol(self) -> str:
        return self.getOrDefault(self.labelCol)


class HasPredictionCol(Params):
    predictionCol = Param(
        "shared", "predictionCol", "prediction column name.", TypeConverters.toString
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(predictionCol="prediction")

    def getPredictionCol(self) -> str:
        return self.getOrDefault(self.predictionCol)


class HasProbabilityCol(Params):
    probabilityCol = Param(
        "shared", "probabilityCol", "class probabilities column name.", TypeConverters.toString
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(probabilityCol="probability")

    def getProbabilityCol(self) -> str:
        return self.getOrDefault(self.probabilityCol)


class HasRawPredictionCol(Params):
    rawPredictionCol = Param(
        "shared", "rawPredictionCol", "raw prediction column name.", TypeConverters.toString
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(rawPredictionCol="rawPrediction")

    def getRawPredictionCol(self) -> str:
        return self.getOrDefault(self.rawPredictionCol)


class HasWeightCol(Params):
    """Instance weights are NOT supported by the native solvers, matching the
    reference (weightCol is None-mapped everywhere; its setters raise,
    e.g. reference clustering.py:357-359). Declared so scripts touching the
    param get the reference's error, not an unknown-param error."""

    weightCol = Param("shared", "weightCol", "instance weight column (unsupported).", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)

    def setWeightCol(self, value: str) -> "HasWeightCol":
        raise ValueError("'weightCol' is not supported (reference parity).")


class HasOutputCol(Params):
    outputCol = Param("shared", "outputCol", "output column name.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)

    def getOutputCol(self) -> str:
        return self.getOrDefault(self.outputCol)


class HasIDCol(Params):
    """Row-id column for algorithms whose output rows must be joined back to
    input rows (kNN, DBSCAN; reference params.py:96-129)."""

    idCol = Param("shared", "idCol", "id column name.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)

    def getIdCol(self) -> str:
        return self.getOrDefault(self.idCol)

    def setIdCol(self, value: str) -> "HasIDCol":
        return self._set(idCol=value)  # type: ignore[return-value]

    def _ensureIdCol(self, df: Any) -> Any:
        """Add a monotonically-increasing row id when idCol is unset
        (reference params.py:106-129)."""
        import numpy as np

        if self.isDefined(self.idCol) and self.getOrDefault(self.idCol) in df.columns:
            return df
        name = self.getOrDefault(self.idCol) if self.isDefined(self.idCol) else "unique_id"
        while name in df.columns:
            name = name + "_"
        self._set(idCol=name)
        from .parallel.context import get_comm

        comm = get_comm()
        n_local = df.num_rows
        counts = comm.allgather_obj(n_local)
        start = sum(counts[: comm.rank])
        return df.with_column(name, np.arange(start, start + n_local, dtype=np.int64))


class HasEnableSparseDataOptim(Params):
    """Sparse-input control (reference params.py:45-66): None = auto-detect,
    True = require sparse CSR path, False = densify."""

    enable_sparse_data_optim = Param(
        "shared",
        "enable_sparse_data_optim",
        "whether to use the sparse CSR data path (None=auto).",
        TypeConverters.toBoolean,
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(enable_sparse_data_optim=None)


class HasVerbose(Params):
    """Native-layer log verbosity (reference params.py:132-159)."""

    verbose = Param(
        "shared",
        "verbose",
        "native log level: bool or int 0-6 (reference params.py:132-159).",
        TypeConverters.identity,
    )

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(verbose=False)


# ---------------------------------------------------------------------------
# Spark-param <-> native-param sync layer
# ---------------------------------------------------------------------------


class _NativeClass:
    """Per-algorithm declaration of the Spark->native param translation
    (reference _CumlClass, params.py:160-260)."""

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        """Spark param name -> native param name; ``None`` -> unsupported
        (error on non-default value); ``""`` -> ignored."""
        return {}

    @classmethod
    def _param_value_mapping(cls) -> Dict[str, Callable[[Any], Union[None, Any]]]:
        """native param name -> lambda mapping the incoming value; returning
        ``None`` flags an unsupported value."""
        return {}

    @classmethod
    def _param_excludes(cls) -> List[str]:
        return []

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        """Default native param dict (reference per-algo _get_cuml_params_default)."""
        return {}


class _NativeParams(_NativeClass, Params):
    """Maintains `native_params` in sync with the Spark-style Params
    (reference _CumlParams, params.py:263-719)."""

    _float32_inputs = True

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._native_params: Dict[str, Any] = dict(self._get_native_params_default())
        self._num_workers: Optional[int] = None
        # reference spark.rapids.ml.cpu.fallback.enabled (params.py:276-285):
        # unsupported params log-and-ignore instead of raising
        from .config import get_conf

        self._fallback_enabled = bool(get_conf("cpu_fallback_enabled"))

    @property
    def native_params(self) -> Dict[str, Any]:
        return self._native_params

    @property
    def cuml_params(self) -> Dict[str, Any]:
        """Reference-compatible alias for native_params (reference params.py:334)."""
        return self._native_params

    @property
    def num_workers(self) -> int:
        """Number of model-parallel workers (= processes = GPUs). Defaults to
        the live communicator's world size (reference params.py:556-588 infers
        from the Spark cluster; here the SPMD world is the cluster)."""
        if self._num_workers is not None:
            return self._num_workers
        from .parallel.context import get_comm

        return get_comm().world_size

    @num_workers.setter
    def num_workers(self, value: int) -> None:
        self._num_workers = value

    def setVerbose(self, value: Union[bool, int]) -> "_NativeParams":
        return self._set_params(verbose=value)  # type: ignore[return-value]

    def _set_params(self: P, **kwargs: Any) -> P:
        """Set Spark params and/or native params, keeping both in sync
        (reference params.py:430-487)."""
        for k, v in kwargs.items():
            if k == "num_workers":
                self._num_workers = v
                continue
            if k == "float32_inputs":
                self._float32_inputs = v
                continue
            if self.hasParam(k):
                self._set(**{k: v})
                self._sync_to_native(k, v)
            elif k in self._native_params:
                # native-only param set directly (reference params.py:474-481)
                if k in ("n_streams", "max_samples_per_batch"):
                    # cuML-parity ctor knobs with no effect in this engine
                    # (HIP kernels don't batch/stream the cuML way)
                    from .config import warn_inert

                    warn_inert(k, type(self).__name__)
                self._native_params[k] = v
            else:
                raise ValueError(f"Unsupported param '{k}'.")
        return self  # type: ignore[return-value]

    def _sync_to_native(self, spark_name: str, value: Any) -> None:
        mapping = self._param_mapping()
        if spark_name not in mapping:
            return
        native_name = mapping[spark_name]
        if native_name is None:
            # unsupported param: error only when set to a non-default value
            default = (
                self._defaultParamMap.get(self.getParam(spark_name))
                if self.hasParam(spark_name)
                else None
            )
            if value != default:
                if self._fallback_enabled:
                    return
                raise ValueError(
                    f"Param '{spark_name}' is not supported on GPU "
                    f"(reference params.py:186-196 semantics)."
                )
            return
        if native_name == "":
            return
        value_mapping = self._param_value_mapping()
        if native_name in value_mapping:
            mapped = value_mapping[native_name](value)
            if mapped is None:
                raise ValueError(
                    f"Value {value!r} for param '{spark_name}' is not supported."
                )
            value = mapped
        if isinstance(native_name, list):
            for nn in native_name:
                self._native_params[nn] = value
        else:
            self._native_params[native_name] = value

    def _sync_all_to_native(self) -> None:
        for p in self.params:
            if self.isSet(p):
                self._sync_to_native(p.name, self.getOrDefault(p))

    def _get_input_columns(self) -> tuple:
        """Resolve (featuresCol, featuresCols) precedence: multi-column wins
        when set (reference params.py:612-637)."""
        features_col: Optional[str] = None
        features_cols: Optional[List[str]] = None
        if isinstance(self, HasFeaturesCols) and self.isSet("featuresCols"):
            features_cols = self.getFeaturesCols()
        elif isinstance(self, HasFeaturesCol) and self.isDefined("featuresCol"):
            features_col = self.getOrDefault("featuresCol")
        else:
            raise ValueError("featuresCol or featuresCols must be set")
        return features_col, features_cols


# Central numeric-validity table for Spark-standard param names, enforced at
# fit() entry (reference behavior: cuML validates and raises per estimator;
# pyspark's ParamValidators reject at set time). (lo, hi, lo_inclusive,
# hi_inclusive); None = unbounded.
_PARAM_BOUNDS = {
    "k": (1, None, True, True),
    "maxIter": (0, None, True, True),
    "initSteps": (1, None, True, True),
    "tol": (0.0, None, True, True),
    "regParam": (0.0, None, True, True),
    "elasticNetParam": (0.0, 1.0, True, True),
    "threshold": (0.0, 1.0, True, True),
    "numTrees": (1, None, True, True),
    "maxDepth": (1, None, True, True),
    "maxBins": (2, 256, True, True),
    "minInstancesPerNode": (1, None, True, True),
    "minInfoGain": (0.0, None, True, True),
    "subsamplingRate": (0.0, 1.0, False, True),
    "eps": (0.0, None, False, True),
    "min_samples": (1, None, True, True),
    "n_neighbors": (1, None, True, True),
    "n_components": (1, None, True, True),
    "n_epochs": (0, None, True, True),
    "numFolds": (2, None, True, True),
    "negative_sample_rate": (1, None, True, True),
    "sample_fraction": (0.0, 1.0, False, True),
}


def validate_param_bounds(inst: "Params") -> None:
    """Raise ValueError for out-of-range numeric params (table above)."""
    for p in inst.params:
        if p.name not in _PARAM_BOUNDS or not inst.isDefined(p.name):
            continue
        v = inst.getOrDefault(p.name)
        if v is None or not isinstance(v, (int, float)):
            continue
        lo, hi, loi, hii = _PARAM_BOUNDS[p.name]
        ok = True
        if lo is not None:
            ok = ok and (v >= lo if loi else v > lo)
        if hi is not None:
            ok = ok and (v <= hi if hii else v < hi)
        if not ok:
            lob = "[" if loi else "("
            hib = "]" if hii else ")"
            raise ValueError(
                f"Param {p.name}={v!r} out of range {lob}{lo}, {hi}{hib}"
            )
