import asyncio
import sys
import os

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cueball_amd.testing import VirtualLoop  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a GPU box (run with -m gpu)")


@pytest.fixture
def vloop():
    """Fresh virtual-time event loop per test."""
    loop = VirtualLoop()
    yield loop
    loop.close()


def run_vt(body):
    """Run an async test body on a fresh VirtualLoop:
    ``run_vt(lambda loop: body(loop))``."""
    loop = VirtualLoop()
    try:
        return loop.run_until_complete(body(loop))
    finally:
        loop.close()
