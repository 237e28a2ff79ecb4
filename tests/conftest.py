"""Shared pytest configuration.

Markers:
  gpu — tests that need a real MI355X; run with ``pytest -m gpu`` on a
        GPU box, excluded in CPU CI via ``-m "not gpu"``.
"""

import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


@pytest.fixture
def tmp_sock(tmp_path):
    """A short unix socket path (AF_UNIX limit is 107 chars)."""
    path = tmp_path / "s.sock"
    return str(path)
