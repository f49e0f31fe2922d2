import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
