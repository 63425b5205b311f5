import os
import sys

import pytest

# Repo root on sys.path so the in-tree packages (min_tfs_client_amd, the
# tensorflow/tensorflow_serving pb2 shims, min_tfs_client alias) import
# without installation.
_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); excluded by "
        "-m 'not gpu'")


def _has_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


@pytest.fixture(scope="session")
def gpu_available():
    return _has_gpu()


def pytest_collection_modifyitems(config, items):
    if _has_gpu():
        return
    skip = pytest.mark.skip(reason="no AMD GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
