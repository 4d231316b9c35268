import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    markexpr = config.getoption("-m", default="")
    if "gpu" in str(markexpr):
        return
    skip = pytest.mark.skip(reason="needs GPU; run with -m gpu")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
