# Copyright (c) Flashy-AMD authors.
import os
import sys
from pathlib import Path

import pytest

# repo root importable (tests run from repo root; keep robust anyway)
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")


@pytest.fixture()
def xp_root(tmp_path, monkeypatch):
    """Redirect the experiment root to a temp dir for the duration of a test."""
    monkeypatch.setenv("_FLASHY_AMD_DIR", str(tmp_path))
    yield tmp_path


@pytest.fixture(autouse=True)
def _reset_current_xp():
    """Tests must not leak the process-global current XP."""
    from flashy_amd import xp as fxp
    yield
    fxp._current_xp = None
