"""Core-framework test via a fake estimator (pattern: reference
tests/test_common_estimator.py — a dummy estimator/model exercising the fit
orchestration, param plumbing, PartitionDescriptor, fitMultiple cache reuse
and persistence without any real solver)."""

from typing import Any, Dict, Optional

import numpy as np
import pytest

from spark_rapids_ml_amd.core import Estimator, Model, _FitContext
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.params import (
    HasFeaturesCol,
    HasFeaturesCols,
    HasPredictionCol,
    Param,
    TypeConverters,
)

from .dist_utils import run_distributed


class _DummyParams(HasFeaturesCol, HasFeaturesCols, HasPredictionCol):
    alpha = Param("dummy", "alpha", "a native float param.", TypeConverters.toFloat)
    k = Param("dummy", "k", "mapped param.", TypeConverters.toInt)
    bad = Param("dummy", "bad", "unsupported (None-mapped).", TypeConverters.toInt)
    ignored = Param("dummy", "ignored", "dropped (''-mapped).", TypeConverters.toInt)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(alpha=1.0, k=2, bad=0, ignored=0)

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        # k maps to native n_widgets; "bad" unsupported (None -> error);
        # "ignored" silently dropped ("" mapping)
        return {"k": "n_widgets", "bad": None, "ignored": "", "alpha": "alpha"}

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        return {"alpha": 1.0, "n_widgets": 2}


class DummyModel(_DummyParams, Model):
    def __init__(self, **attrs: Any) -> None:
        super().__init__(**attrs)
        self._set_params()

    @property
    def mean_(self) -> np.ndarray:
        return np.asarray(self._model_attributes["mean_"])

    def _transform_array(self, X: Any):
        # distance of each row to the fitted global mean
        return np.linalg.norm(np.asarray(X) - self.mean_[None, :], axis=1)


class DummyEstimator(_DummyParams, Estimator):
    """Fits the global column mean via one fused all-reduce; records what the
    fit hook observed so tests can assert on the orchestration."""

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._set_params(**kwargs)
        self.seen: Dict[str, Any] = {}

    def _fit_array(self, X, y, ctx: _FitContext, params: Dict[str, Any]):
        import torch

        Xt = ctx.device_tensor(X, dtype=torch.float64)
        self.seen.setdefault("tensor_ids", []).append(id(Xt))
        self.seen["params"] = dict(params)
        self.seen["pdesc_m"] = ctx.pdesc.m
        self.seen["pdesc_parts"] = list(ctx.pdesc.parts_rank_size)
        buf = torch.cat([Xt.sum(dim=0), torch.tensor([float(Xt.shape[0])], dtype=Xt.dtype)])
        buf = ctx.comm.allreduce_t(buf)
        mean = (buf[:-1] / buf[-1]).cpu().numpy()
        return {"mean_": mean * float(params["alpha"])}

    def _create_model(self, attrs):
        return DummyModel(**attrs)


def _df(n=40, d=3, seed=0):
    rng = np.random.default_rng(seed)
    return rng.normal(size=(n, d)).astype(np.float64)


def test_dummy_fit_and_transform():
    X = _df()
    est = DummyEstimator(k=5)
    model = est.fit(DataFrame.from_numpy(X))
    assert np.allclose(model.mean_, X.mean(axis=0))
    # param mapping reached the fit hook
    assert est.seen["params"]["n_widgets"] == 5
    assert est.seen["pdesc_m"] == 40
    out = model.transform(DataFrame.from_numpy(X))
    expect = np.linalg.norm(X - X.mean(axis=0), axis=1)
    assert np.allclose(np.asarray(out["prediction"]), expect)


def test_dummy_unsupported_param_policy():
    with pytest.raises(ValueError):
        DummyEstimator(bad=1)  # None-mapped -> error
    est = DummyEstimator(ignored=7)  # ""-mapped -> silently dropped
    assert "ignored" not in est._native_params
    assert est.getOrDefault("ignored") == 7  # still recorded as a Spark param


def test_dummy_fit_multiple_shares_one_ingest():
    X = _df()
    est = DummyEstimator()
    maps = [{est.getParam("alpha"): 1.0}, {est.getParam("alpha"): 2.0}]
    models = dict(est.fitMultiple(DataFrame.from_numpy(X), maps))
    assert np.allclose(models[1].mean_, 2.0 * models[0].mean_)
    # the device ingest ran ONCE: both param maps hit the _FitContext cache
    assert len(set(est.seen["tensor_ids"])) == 1
    assert len(est.seen["tensor_ids"]) == 2


def test_dummy_persistence_roundtrip(tmp_model_path):
    X = _df()
    model = DummyEstimator(k=9, alpha=1.5).fit(DataFrame.from_numpy(X))
    model.save(tmp_model_path)
    loaded = DummyModel.load(tmp_model_path)
    assert np.allclose(loaded.mean_, model.mean_)
    assert loaded.getOrDefault("k") == 9
    assert loaded._native_params["n_widgets"] == 9


def test_dummy_multicol_input():
    X = _df(d=3)
    df = DataFrame({"a": X[:, 0], "b": X[:, 1], "c": X[:, 2]})
    model = DummyEstimator(featuresCols=["a", "b", "c"]).fit(df)
    assert np.allclose(model.mean_, X.mean(axis=0), atol=1e-6)


def _dist_dummy(_):
    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    X = _df(n=50, seed=1)
    shard = X[comm.rank :: comm.world_size]
    est = DummyEstimator()
    model = est.fit(DataFrame.from_numpy(shard))
    return model.mean_, est.seen["pdesc_m"], est.seen["pdesc_parts"]


def test_dummy_distributed_partition_descriptor():
    results = run_distributed(_dist_dummy, world_size=2, args=(None,))
    X = _df(n=50, seed=1)
    for mean, m, parts in results:
        assert np.allclose(mean, X.mean(axis=0))
        assert m == 50
        assert sorted(parts) == [(0, 25), (1, 25)]
