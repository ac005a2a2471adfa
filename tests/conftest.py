import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    # Auto-skip gpu tests when no HIP device is present and -m gpu not requested
    markexpr = config.getoption("-m") or ""
    if "gpu" in markexpr:
        return
    skip_gpu = pytest.mark.skip(reason="needs MI355X GPU (-m gpu)")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
