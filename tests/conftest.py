import asyncio
import sys
from pathlib import Path

import pytest

# repo root on sys.path so `import gpu_provisioner_amd` works without install
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def run(coro, timeout: float = 60.0):
    """Run an async test body on a fresh event loop with a safety timeout.
    On timeout, every live task's stack is dumped first — an idle-parked
    loop with unfinished work is a lost-wakeup bug, and the parked frames
    are the evidence."""

    async def wrapped():
        inner = asyncio.ensure_future(coro)
        try:
            return await asyncio.wait_for(asyncio.shield(inner), timeout)
        except asyncio.TimeoutError:
            import traceback

            print(f"\n=== run() timeout after {timeout}s: task dump ===")
            for task in asyncio.all_tasks():
                if task is asyncio.current_task():
                    continue
                print(f"--- {task.get_name()} done={task.done()}")
                for frame in task.get_stack(limit=4):
                    traceback.print_stack(frame, limit=1)
            inner.cancel()
            try:
                await inner
            except Exception:
                pass
            raise

    return asyncio.run(wrapped())


@pytest.fixture
def anyio_backend():
    return "asyncio"
