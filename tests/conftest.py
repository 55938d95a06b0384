import os
import sys

import pytest

os.environ.setdefault("OMP_NUM_THREADS", "8")

def _cap_torch_threads():
    import torch

    torch.set_num_threads(min(8, os.cpu_count() or 8))

_cap_torch_threads()

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires a ROCm GPU")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config, items):
    import torch

    if not torch.cuda.is_available():
        skip_gpu = pytest.mark.skip(reason="no GPU available")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip_gpu)


@pytest.fixture
def tmp_model_path(tmp_path):
    return str(tmp_path / "model")
