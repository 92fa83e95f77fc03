import os
import sys

import pytest

# make the repo root importable regardless of invocation directory
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); skipped in CPU CI")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no HIP device is present,
    unless the run explicitly selects them (-m gpu)."""
    if config.getoption("-m") and "gpu" in config.getoption("-m"):
        return
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
