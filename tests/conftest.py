import os
import sys

import pytest
import torch

# repo root on sys.path so `import lstm_tensorspark_amd` works from tests/
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on the MI355X box)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def cpu_seed():
    torch.manual_seed(1234)
    return 1234
