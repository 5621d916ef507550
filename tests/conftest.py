import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present, unless
    the user explicitly selected them with -m gpu."""
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    markexpr = config.getoption("-m", default="")
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords and "gpu" not in str(markexpr):
            item.add_marker(skip)
