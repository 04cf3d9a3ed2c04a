import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (skipped on CPU-only machines)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def gpu():
    assert torch.cuda.is_available()
    import flink_ms_amd.ops as ops
    assert ops.hip_available(), "HIP extension must be built on GPU boxes"
    return torch.device("cuda:0")
