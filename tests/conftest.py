"""Shared fixtures.

No pytest-asyncio in this image: async tests run their coroutine via the
``run_async`` helper (fresh event loop per test).
"""

from __future__ import annotations

import asyncio
import contextlib
from typing import Any, Awaitable, Callable, Tuple

import pytest

from llmq_amd.core.config import Config


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")
    config.addinivalue_line("markers", "integration: spins up a live in-process broker")
    config.addinivalue_line("markers", "slow: long-running")


def run_async(coro: Awaitable[Any], timeout: float = 60.0) -> Any:
    async def wrapped():
        return await asyncio.wait_for(coro, timeout=timeout)

    return asyncio.run(wrapped())


@pytest.fixture()
def anyio_run():
    return run_async


@contextlib.asynccontextmanager
async def live_broker(data_dir=None, max_retries: int = 3):
    """An in-process broker on an ephemeral port + a Config pointing at it."""
    from llmq_amd.broker.server import BrokerServer

    server = BrokerServer("127.0.0.1", 0, data_dir=str(data_dir) if data_dir else None,
                          max_retries=max_retries)
    await server.serve()
    config = Config(broker_url=f"llmq://127.0.0.1:{server.port}")
    try:
        yield server, config
    finally:
        await server.close()


@pytest.fixture()
def broker_ctx():
    return live_broker


@pytest.fixture()
def sample_job_dict():
    return {"id": "job-001", "prompt": "Translate: {text}", "text": "hallo wereld"}
