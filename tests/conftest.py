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


# The driver runs the GPU suite with -x: run the foundational parity gates
# (scan parity, batch, hnsw search) BEFORE the newer/riskier suites so one
# regression cannot leave the core evidence unobserved.
_GPU_FILE_ORDER = [
    "test_gpu_parity.py",   # single-query scan parity gates
    "test_gpu_batch.py",    # batched MFMA path parity
    "test_gpu_hnsw.py",     # graph search parity + recall bars
    "test_gpu_index.py",    # index layer (writes/pendings/filtered)
    "test_gpu_insert.py",   # GPU-accelerated build
]


def pytest_collection_modifyitems(config, items):
    def key(item):
        fname = os.path.basename(str(item.fspath))
        try:
            rank = _GPU_FILE_ORDER.index(fname)
        except ValueError:
            rank = len(_GPU_FILE_ORDER)
        return rank

    items.sort(key=key)
