import asyncio

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is visible."""
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def run():
    """Run a coroutine to completion on a fresh event loop."""
    def _run(coro, timeout=30.0):
        loop = asyncio.get_event_loop_policy().new_event_loop()
        try:
            return loop.run_until_complete(asyncio.wait_for(coro, timeout))
        finally:
            pending = asyncio.all_tasks(loop)
            for t in pending:
                t.cancel()
            if pending:
                # bounded: a task that swallows cancellation must not hang
                # the whole suite
                loop.run_until_complete(asyncio.wait_for(
                    asyncio.gather(*pending, return_exceptions=True), 15))
            loop.close()
    return _run
