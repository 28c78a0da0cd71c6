"""Result collection → stdout JSONL (reference parity: llmq/cli/receive.py:17-303).

Durable + resumable: results stay queued until acked, and a result is acked
only after it has been written and flushed to stdout — re-running resumes
where the last run stopped (reference receive.py:112-125).
Exits after ``timeout`` seconds with no new result.
"""

from __future__ import annotations

import asyncio
import logging
import signal
import sys
import time
from typing import Any, Optional

from llmq_amd.core.client import BrokerClient, Delivery
from llmq_amd.core.config import get_config
from llmq_amd.core.models import Result
from llmq_amd.core.pipeline import PipelineConfig

logger = logging.getLogger(__name__)


class ResultReceiver:
    def __init__(
        self,
        queue_name: str,
        timeout: float = 30.0,
        limit: Optional[int] = None,
        skip_filtered: bool = False,
    ):
        self.queue_name = queue_name
        self.timeout = timeout
        self.limit = limit
        self.skip_filtered = skip_filtered
        self.config = get_config()
        self.client = BrokerClient(self.config)
        self.received = 0
        self._last_seen = time.time()
        self._stop = asyncio.Event()

    def _handle_signal(self, *_a: Any) -> None:
        self._stop.set()

    async def _on_result(self, delivery: Delivery) -> None:
        try:
            result = Result.model_validate_json(delivery.body)
        except Exception as exc:
            logger.warning("dropping unparseable result: %s", exc)
            await delivery.nack(requeue=False, error=f"bad result: {exc}")
            return
        self._last_seen = time.time()
        if self.limit is not None and self.received >= self.limit:
            # over-delivered (prefetch window): leave it queued for the next
            # `llmq receive` — results are durable/resumable
            await delivery.nack(requeue=True)
            self._stop.set()
            return
        if self.skip_filtered and getattr(result, "filtered", False):
            await delivery.ack()
            return
        sys.stdout.write(delivery.body.rstrip("\n") + "\n")
        sys.stdout.flush()
        await delivery.ack()
        self.received += 1
        if self.limit is not None and self.received >= self.limit:
            self._stop.set()

    async def run(self) -> int:
        loop = asyncio.get_event_loop()
        for sig in (signal.SIGINT, signal.SIGTERM):
            try:
                loop.add_signal_handler(sig, self._handle_signal)
            except (NotImplementedError, RuntimeError):
                pass
        await self.client.connect()
        await self.client.consume_results(self.queue_name, self._on_result, prefetch=1000)
        start = time.time()
        self._last_seen = start
        while not self._stop.is_set():
            try:
                await asyncio.wait_for(self._stop.wait(), timeout=0.5)
            except asyncio.TimeoutError:
                pass
            if time.time() - self._last_seen > self.timeout:
                break
        elapsed = max(time.time() - start, 1e-9)
        print(
            f"Received {self.received} results in {elapsed:.1f}s "
            f"({self.received / elapsed:.1f} results/s)",
            file=sys.stderr,
        )
        await self.client.disconnect()
        return self.received


class PipelineResultReceiver(ResultReceiver):
    def __init__(self, pipeline: PipelineConfig, **kwargs):
        # consume_results appends ".results"; hand it the bare prefix
        base = pipeline.get_pipeline_results_queue_name().removesuffix(".results")
        super().__init__(base, **kwargs)
        self.pipeline = pipeline


def run_receive(queue_name: str, timeout: float, limit: Optional[int], skip_filtered: bool = False) -> None:
    receiver = ResultReceiver(queue_name, timeout=timeout, limit=limit, skip_filtered=skip_filtered)
    asyncio.run(receiver.run())


def run_pipeline_receive(pipeline_path: str, timeout: float, limit: Optional[int]) -> None:
    pipeline = PipelineConfig.from_yaml_file(pipeline_path)
    receiver = PipelineResultReceiver(pipeline, timeout=timeout, limit=limit)
    asyncio.run(receiver.run())
