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


def test_standardize_dataset_matches_numpy():
    import torch

    from spark_rapids_ml_amd.parallel.context import PartitionDescriptor, get_comm
    from spark_rapids_ml_amd.utils import standardize_dataset

    rng = np.random.default_rng(0)
    X = rng.normal(loc=3.0, scale=2.5, size=(200, 6))
    comm = get_comm()
    pdesc = PartitionDescriptor.build(comm, X.shape[0], X.shape[1])
    Xs, mean, std = standardize_dataset(torch.from_numpy(X), comm, pdesc)
    assert np.allclose(mean.numpy(), X.mean(axis=0))
    assert np.allclose(std.numpy(), X.std(axis=0, ddof=1))
    assert np.allclose(Xs.numpy().mean(axis=0), 0.0, atol=1e-12)


def test_torch_ref_logistic_oracle_consistency():
    """The torch_ref logistic oracle and the production glm pass agree
    (gradient and loss) on the same weights."""
    import torch

    from spark_rapids_ml_amd.ops.glm import logistic_grad_loss
    from spark_rapids_ml_amd.ops.torch_ref import logistic_forward_grad

    rng = np.random.default_rng(0)
    X = torch.from_numpy(rng.normal(size=(100, 5)))
    y = torch.from_numpy(rng.integers(0, 3, size=100))
    W = torch.from_numpy(rng.normal(size=(3, 6)) * 0.1)
    g1, l1 = logistic_forward_grad(X, y, W, fit_intercept=True)
    g2, l2 = logistic_grad_loss(X, y, W, fit_intercept=True)
    assert np.allclose(g1.numpy(), g2.numpy(), atol=1e-10)
    assert np.isclose(float(l1), float(l2), atol=1e-10)


def _dist_dead_rank(_):
    """Rank 1 dies before the collective; rank 0 must fail FAST (bounded by
    SRML_COMM_TIMEOUT_S), not hang — the reference's all-or-nothing barrier
    + NCCL-abort semantics (cuml_context.py:162-167)."""
    import os
    import time

    import torch

    from spark_rapids_ml_amd.parallel.context import get_comm

    comm = get_comm()
    if comm.rank == 1:
        os._exit(0)  # simulate a dead worker (no clean shutdown)
    t0 = time.perf_counter()
    try:
        comm.allreduce(torch.ones(4))
        return ("no-error", time.perf_counter() - t0)
    except Exception as e:
        return ("error", time.perf_counter() - t0, type(e).__name__)


def test_dead_rank_fails_fast(monkeypatch):
    import pytest as _pytest

    from .dist_utils import run_distributed

    monkeypatch.setenv("SRML_COMM_TIMEOUT_S", "8")
    with _pytest.raises(RuntimeError):
        # rank 1 exits without reporting; the harness reports the missing
        # rank (and rank 0's collective error) as a failure
        run_distributed(_dist_dead_rank, world_size=2, args=(None,))
