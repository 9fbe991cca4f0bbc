import os
import sys

import pytest
import torch

# repo root on sys.path so `import sac` / `torch_actor_critic_amd` work
# when pytest is invoked from anywhere
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu)")
    config.addinivalue_line(
        "markers", "slow: long-running CPU test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def _deterministic_seed():
    torch.manual_seed(0)
    yield
