import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture(autouse=True)
def _deterministic_seed():
    import torch

    torch.manual_seed(0)
    yield
