import asyncio
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require a real MI355X GPU box")
    # Fresh checkouts have no in-tree _amcore.so (it is git-ignored); build it
    # once so the suite exercises the native hot path everywhere a toolchain
    # exists. Failure falls back to the pure-Python path (tests that REQUIRE
    # native will then fail loudly, which is the point on a GPU box).
    try:
        import active_monitor_amd._amcore  # noqa: F401
    except ImportError:
        subprocess.run(
            [sys.executable, "setup.py", "build_ext", "--inplace"],
            cwd=REPO, capture_output=True, timeout=300,
        )


@pytest.fixture
def run():
    """Run a coroutine to completion on a fresh event loop."""

    def _run(coro, timeout=60.0):
        return asyncio.run(asyncio.wait_for(coro, timeout))

    return _run
