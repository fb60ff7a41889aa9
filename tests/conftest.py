import os
import sys

import pytest

# repo root on sys.path so `oracle`, `denormalized_amd`, `tests.pyref` import
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    # Nothing automatic: gpu tests are explicitly marked. But guard against
    # accidentally running them where no GPU exists unless explicitly selected.
    if config.getoption("-m") and "gpu" in config.getoption("-m"):
        return
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if not has_gpu:
        skip = pytest.mark.skip(reason="no GPU in this environment")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
