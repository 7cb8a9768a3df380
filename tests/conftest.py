import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X); run with -m gpu")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        # the torch reference arms of parity tests otherwise go through
        # hipBLASLt, whose per-process algorithm choice varies run to run
        # (observed: rare tolerance flakes in fused-vs-torch comparisons,
        # chaotic trajectories amplifying the different rounding); rocBLAS
        # picks deterministically per shape
        try:
            torch.backends.cuda.preferred_blas_library("cublas")
        except Exception:
            pass
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
