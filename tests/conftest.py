import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run with -m gpu)")
    config.addinivalue_line("markers", "slow: long-running CPU test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
