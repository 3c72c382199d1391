import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")
    config.addinivalue_line(
        "markers", "slow: long-running contract test (8-rank subprocess)")


def pytest_collection_modifyitems(config, items):
    if not torch.cuda.is_available():
        skip = pytest.mark.skip(reason="no GPU available")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
