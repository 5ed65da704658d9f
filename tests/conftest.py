import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) and the HIP extension"
    )


@pytest.fixture
def exp_dir(tmp_path, monkeypatch):
    """Fresh experiment base dir + fresh Environment singleton per test."""
    import maggy_amd.core.environment as envmod

    monkeypatch.setenv("MAGGY_LOG_DIR", str(tmp_path))
    envmod.Environment.set_instance(envmod.Environment(base_dir=str(tmp_path)))
    yield str(tmp_path)
    envmod.Environment.set_instance(None)
