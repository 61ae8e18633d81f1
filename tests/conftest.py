import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X (run with `pytest -m gpu` on the GPU box)")


def pytest_collection_modifyitems(config, items):
    markexpr = config.getoption("-m", default="")
    if "gpu" in str(markexpr):
        return
    skip_gpu = pytest.mark.skip(reason="needs GPU (run with -m gpu)")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
