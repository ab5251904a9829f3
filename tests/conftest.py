import sys
from pathlib import Path

import pytest  # noqa: F401

# repo root on sys.path so `oracle` and `magi_attention` import without install
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test needs a real MI355X GPU (run via gpurun)"
    )
