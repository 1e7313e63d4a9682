import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )


@pytest.fixture(autouse=True)
def _isolated_home(tmp_path, monkeypatch):
    """Keep ~/.bee2bee writes inside the test sandbox."""
    monkeypatch.setenv("BEE2BEE_HOME", str(tmp_path / "bee2bee_home"))
    yield
