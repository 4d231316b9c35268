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


@pytest.fixture(scope="session", autouse=True)
def _torch_hip_context_first(request):
    """Initialize torch's HIP context before any test touches the arroyo-amd
    library.  torch's wheel bundles its own HIP runtime; if the system-ROCm
    runtime our .so links initializes the device first, torch's later
    _cuda_init fails with "No HIP GPUs are available".  Initializing torch
    first works in either direction (bench.py relies on the same order)."""
    if "gpu" in str(request.config.getoption("-m", default="")):
        try:
            import torch
            if torch.cuda.is_available():
                torch.cuda.init()
        except Exception:
            pass
