import asyncio

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on the GPU box)")
    config.addinivalue_line("markers", "slow: long-running test")


@pytest.fixture
def event_loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def run_async(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


@pytest.fixture
def runner():
    return run_async
