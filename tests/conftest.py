import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require a ROCm GPU (run on MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def device():
    return torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
