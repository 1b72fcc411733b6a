import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture(autouse=True)
def _cpu_default_for_cpu_tests(request):
    """Non-gpu tests must behave identically on a GPU box: paddle's
    default device follows CUDA availability, so pin it to CPU unless
    the test is @gpu-marked (a CPU-built model otherwise lands on the
    GPU and mixes devices with CPU test inputs)."""
    if "gpu" in request.keywords or not torch.cuda.is_available():
        yield
        return
    import paddle_amd as paddle
    from paddle_amd import framework as fw
    prev = fw._default_device
    paddle.set_device("cpu")
    try:
        yield
    finally:
        fw._default_device = prev


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
