import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X (HIP) device; run with -m gpu on a GPU box"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no HIP device in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
