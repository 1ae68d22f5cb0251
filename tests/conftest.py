import asyncio

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require a real MI355X GPU box")


@pytest.fixture
def run():
    """Run a coroutine to completion on a fresh event loop."""

    def _run(coro, timeout=60.0):
        return asyncio.run(asyncio.wait_for(coro, timeout))

    return _run
