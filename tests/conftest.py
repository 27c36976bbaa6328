import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# The simulated-multi-rank comm tests run 8 spin-waiting kernels on 8
# streams of ONE process; ROCm's default of 4 hardware queues would
# serialize them into a deadlock.  Must be set before HIP runtime init
# (conftest imports before any CUDA call).  Real TP runs one process
# per GPU and needs no such setting.
os.environ.setdefault("GPU_MAX_HW_QUEUES", "10")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except ImportError:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
