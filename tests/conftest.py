import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE_ROOT = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def reference_root():
    if not os.path.isdir(REFERENCE_ROOT):
        pytest.skip("reference corpus not mounted on this box")
    return REFERENCE_ROOT
