import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (ROCm) GPU")
    config.addinivalue_line(
        "markers", "mpi: test spawns multiple torch.distributed ranks")


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(0)
    yield
