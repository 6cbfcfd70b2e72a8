import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run via gpurun)")


@pytest.fixture(autouse=True)
def _clean_env(monkeypatch):
    # Tests control their own world; don't inherit a torchrun environment.
    for var in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MLSL_RANK", "MLSL_SIZE"):
        monkeypatch.delenv(var, raising=False)
    yield
