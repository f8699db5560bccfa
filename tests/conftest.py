import asyncio
import sys
from pathlib import Path

import pytest

# repo root on sys.path so `import gpu_provisioner_amd` works without install
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def run(coro, timeout: float = 60.0):
    """Run an async test body on a fresh event loop with a safety timeout."""
    async def wrapped():
        return await asyncio.wait_for(coro, timeout)

    return asyncio.run(wrapped())


@pytest.fixture
def anyio_backend():
    return "asyncio"
