import os
import socket
import sys
from pathlib import Path

import pytest

# repo root importable (tests run from anywhere)
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); skipped on CPU-only")


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(skip_gpu)


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture
def dist_env():
    """Environment for single-node torch.distributed rendezvous."""
    return {
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(free_port()),
    }
