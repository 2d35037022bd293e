import os
import sys

import pytest

# repo root on sys.path so `import amdtrain` works without install
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("OMP_NUM_THREADS", "2")
os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    import torch
    # global hang insurance (pytest-timeout): no single test may eat the
    # CI/driver budget; multi-process launch tests get headroom
    for item in items:
        if item.get_closest_marker("timeout") is None:
            item.add_marker(pytest.mark.timeout(600))
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
