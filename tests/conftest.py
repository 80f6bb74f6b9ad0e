import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")


def _has_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


def pytest_runtest_setup(item):
    if "gpu" in [m.name for m in item.iter_markers()]:
        if not _has_gpu():
            pytest.skip("no GPU in this environment")
