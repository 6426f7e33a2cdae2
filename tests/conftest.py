import os
import sys

import pytest

# Make the repo root importable regardless of where pytest is invoked from.
_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run with -m gpu on an MI355X box)")


def pytest_collection_modifyitems(config, items):
    # When no GPU is present and the user didn't filter, auto-skip gpu tests
    # so a plain `pytest tests/` run stays green on CPU-only machines.
    if config.getoption("-m"):
        return
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU present")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
