import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def tmp_socket_dir(tmp_path):
    """Short socket dir (unix socket paths are capped at ~107 chars)."""
    d = tmp_path / "dp"
    d.mkdir()
    return str(d)
