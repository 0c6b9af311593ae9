import os
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(REPO))

# keep CPU tests fast and deterministic
os.environ.setdefault("OMP_NUM_THREADS", "4")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
