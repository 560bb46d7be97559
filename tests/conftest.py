import sys
from pathlib import Path

import pytest

# Make the repo root importable without installation.
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


@pytest.fixture
def anyio_backend():
    return "asyncio"
