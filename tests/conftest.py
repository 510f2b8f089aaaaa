import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def small_geo():
    from quda_amd import LatticeGeometry
    return LatticeGeometry((4, 4, 4, 4))


@pytest.fixture
def rect_geo():
    from quda_amd import LatticeGeometry
    return LatticeGeometry((4, 6, 2, 8))
