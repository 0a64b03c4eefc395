import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def small3():
    """3-mode tensor with duplicates and an empty slice — repair-path food."""
    import splatt_amd as sp
    return sp.SpTensor.synthetic([50, 40, 60], 5000, seed=7)


@pytest.fixture
def med4():
    import splatt_amd as sp
    return sp.SpTensor.synthetic([30, 25, 40, 20], 20000, seed=11)


@pytest.fixture
def med5():
    import splatt_amd as sp
    return sp.SpTensor.synthetic([15, 20, 25, 10, 12], 15000, seed=13)
