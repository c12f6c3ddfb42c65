import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        # box-safety: no single GPU test may hang the suite (pytest-
        # timeout dumps all stacks and kills the run at the cap, far
        # below gpurun's limit)
        for item in items:
            if "gpu" in item.keywords and \
                    item.get_closest_marker("timeout") is None:
                item.add_marker(pytest.mark.timeout(240,
                                                    method="thread"))
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
