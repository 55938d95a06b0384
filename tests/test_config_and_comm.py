"""Config tier + communicator unit tests."""

import numpy as np
import torch

from spark_rapids_ml_amd.config import get_conf, reset_conf, set_conf
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.parallel.context import get_comm


def test_conf_defaults_and_set():
    reset_conf()
    assert get_conf("cpu_fallback_enabled") is False
    set_conf("cpu_fallback_enabled", True)
    assert get_conf("cpu_fallback_enabled") is True
    reset_conf()


def test_conf_env(monkeypatch):
    reset_conf()
    monkeypatch.setenv("SRML_VERBOSE", "true")
    assert get_conf("verbose") is True


def test_cpu_fallback_allows_unsupported_param():
    from spark_rapids_ml_amd import KMeans

    reset_conf()
    set_conf("cpu_fallback_enabled", True)
    try:
        km = KMeans(weightCol="w")  # would raise without fallback
        assert km.getOrDefault("weightCol") == "w"
    finally:
        reset_conf()


def test_allreduce_t_single_process():
    comm = get_comm()
    t = torch.arange(4, dtype=torch.float64)
    out = comm.allreduce_t(t.clone())
    assert torch.equal(out, t)


def test_random_split_and_sample():
    df = DataFrame({"a": np.arange(1000)})
    tr, te = df.randomSplit([0.8, 0.2], seed=1)
    assert tr.num_rows + te.num_rows == 1000
    assert 700 < tr.num_rows < 900
    s = df.sample(0.1, seed=2)
    assert 50 < s.num_rows < 200
