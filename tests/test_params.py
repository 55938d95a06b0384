"""Param system semantics (model: reference tests/test_common_estimator.py's
param-mapping checks)."""

import numpy as np
import pytest

from spark_rapids_ml_amd import KMeans, LinearRegression
from spark_rapids_ml_amd.params import Param, Params, TypeConverters


def test_param_identity_and_defaults():
    km = KMeans()
    assert km.hasParam("k")
    assert km.getOrDefault("k") == 2
    assert not km.isSet("k")
    km.setK(5)
    assert km.isSet("k")
    assert km.getOrDefault("k") == 5
    assert km.native_params["n_clusters"] == 5


def test_param_sync_spark_to_native():
    km = KMeans(k=7, maxIter=13, tol=1e-3, seed=11)
    assert km.native_params["n_clusters"] == 7
    assert km.native_params["max_iter"] == 13
    assert km.native_params["tol"] == 1e-3
    assert km.native_params["random_state"] == 11


def test_native_only_param_accepted():
    # native-only params accepted in constructors (reference README:157-161)
    km = KMeans(oversampling_factor=3.0)
    assert km.native_params["oversampling_factor"] == 3.0


def test_unsupported_param_value_errors():
    with pytest.raises(ValueError):
        KMeans(initMode="bogus")


def test_unsupported_param_none_mapping_errors():
    # weightCol maps to None -> error on set (reference params.py:186-196)
    with pytest.raises(ValueError):
        KMeans(weightCol="w")


def test_unknown_param_errors():
    with pytest.raises(ValueError):
        KMeans(definitely_not_a_param=1)


def test_copy_isolates_maps():
    km = KMeans(k=3)
    km2 = km.copy({"k": 9} and {km.getParam("k"): 9})
    assert km2.getOrDefault("k") == 9
    assert km.getOrDefault("k") == 3


def test_explain_params_runs():
    s = KMeans().explainParams()
    assert "maxIter" in s


def test_value_mapping_initmode():
    km = KMeans(initMode="random")
    assert km.native_params["init"] == "random"


def test_linreg_param_mapping():
    lr = LinearRegression(regParam=0.5, elasticNetParam=0.3, maxIter=7)
    assert lr.native_params["alpha"] == 0.5
    assert lr.native_params["l1_ratio"] == 0.3
    assert lr.native_params["max_iter"] == 7
