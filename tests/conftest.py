import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) and the HIP "
        "extension; run with `pytest -m gpu` on a GPU box")


@pytest.fixture
def tmp_system_path(tmp_path, monkeypatch):
    """Point the index system path at a temp dir."""
    p = tmp_path / "indexes"
    p.mkdir()
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(p))
    return str(p)
