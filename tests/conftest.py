"""Test configuration.

Markers:
  gpu - requires a real MI355X (run via gpurun / driver round-end)
"""
import os
import sys
import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")


def _ensure_built(lib: Path, makedir: Path):
    if not lib.exists():
        subprocess.run(["make", "-C", str(makedir)], check=True, capture_output=True)


@pytest.fixture(scope="session")
def orc():
    """ctypes handle to the CPU oracle (test infrastructure only)."""
    from tests.orc_bindings import OracleLib
    _ensure_built(REPO / "oracle" / "liborc.so", REPO / "oracle")
    return OracleLib(str(REPO / "oracle" / "liborc.so"))
