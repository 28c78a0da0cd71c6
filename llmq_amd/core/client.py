"""Broker client — the reference BrokerManager's role (llmq/core/broker.py:18-353)
against the in-tree broker instead of AMQP.

Method surface parity:
  connect / disconnect               (broker.py:27-55, retry ×5 expo backoff)
  setup_queue_infrastructure         (broker.py:57-81 — declares <q> and <q>.results)
  setup_pipeline_infrastructure      (broker.py:83-113 — per-stage queues + results)
  publish_job / publish_result       (broker.py:115-143)
  consume_jobs / consume_results     (broker.py:195-220)
  get_queue_stats                    (broker.py:222-289; stats come from the broker
                                      itself — no separate management HTTP API needed)
  get_failed_messages / clear_queue  (broker.py:291-353; .failed here is a real DLQ)
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, Awaitable, Callable, Dict, List, Optional

from llmq_amd.broker import protocol
from llmq_amd.broker.server import FAILED_SUFFIX, RESULTS_SUFFIX
from llmq_amd.core.config import Config, get_config
from llmq_amd.core.models import ErrorInfo, Job, QueueStats, Result
from llmq_amd.core.pipeline import PipelineConfig

logger = logging.getLogger(__name__)

DeliverCallback = Callable[["Delivery"], Awaitable[None]]


class Delivery:
    """One in-flight message. Call exactly one of ack()/nack()."""

    __slots__ = ("client", "queue", "tag", "body", "redelivered", "attempts", "_done")

    def __init__(self, client: "BrokerClient", frame: Dict[str, Any]):
        self.client = client
        self.queue = frame["queue"]
        self.tag = frame["tag"]
        self.body = frame["body"]
        self.redelivered = frame.get("redelivered", False)
        self.attempts = frame.get("attempts", 1)
        self._done = False

    async def ack(self) -> None:
        if not self._done:
            self._done = True
            await self.client._send({"m": "ack", "tag": self.tag})

    async def nack(self, requeue: bool = True, error: str = "", worker: str = "") -> None:
        if not self._done:
            self._done = True
            await self.client._send(
                {"m": "nack", "tag": self.tag, "requeue": requeue, "error": error, "worker": worker}
            )


AMQP_CLIENT_CLS = None  # set by llmq_amd.core.amqp_client on import


class BrokerClient:
    """Factory + default (in-tree JSON protocol) backend.

    ``BrokerClient(config)`` returns the AMQP 0-9-1 backend
    (core/amqp_client.py) when the broker URL scheme is ``amqp://`` /
    ``amqps://`` — the reference's RabbitMQ deployment story — and this
    class for ``llmq://`` (the in-tree broker, which itself also answers
    AMQP on the same port)."""

    def __new__(cls, config: Optional[Config] = None):
        if cls is BrokerClient:
            cfg = config or get_config()
            if cfg.broker_url.split("://", 1)[0].lower() in ("amqp", "amqps"):
                global AMQP_CLIENT_CLS
                if AMQP_CLIENT_CLS is None:
                    from llmq_amd.core import amqp_client  # noqa: F401 — registers
                return super().__new__(AMQP_CLIENT_CLS)
        return super().__new__(cls)

    def __init__(self, config: Optional[Config] = None):
        self.config = config or get_config()
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._read_task: Optional[asyncio.Task] = None
        self._pending: Dict[int, asyncio.Future] = {}
        self._next_i = 0
        # One consumer callback per queue per client, registered BEFORE the
        # consume request is sent (a delivery can race ahead of the reply).
        self._queue_callbacks: Dict[str, DeliverCallback] = {}
        self._consumer_queue: Dict[int, str] = {}  # consumer_id -> queue
        self._send_lock = asyncio.Lock()
        self._cb_tasks: set = set()  # strong refs: bare ensure_future may be GC'd
        self._closed = True
        self.default_prefetch = self.config.queue_prefetch

    @property
    def connected(self) -> bool:
        return not self._closed and self._writer is not None

    # -- connection ------------------------------------------------------

    async def connect(self, retries: int = 5) -> None:
        delay = 0.5
        last_exc: Optional[Exception] = None
        for attempt in range(retries):
            try:
                self._reader, self._writer = await asyncio.open_connection(
                    self.config.broker_host,
                    self.config.broker_port,
                    limit=protocol.MAX_FRAME,
                )
                self._closed = False
                self._read_task = asyncio.create_task(self._read_loop())
                await self.call({"m": "ping"})
                return
            except (ConnectionError, OSError, asyncio.TimeoutError) as exc:
                last_exc = exc
                logger.warning(
                    "broker connect attempt %d/%d failed: %s", attempt + 1, retries, exc
                )
                await asyncio.sleep(delay)
                delay = min(delay * 2, 8.0)
        raise ConnectionError(
            f"Could not connect to broker at {self.config.broker_url}: {last_exc}"
        )

    async def disconnect(self) -> None:
        self._closed = True
        if self._read_task:
            self._read_task.cancel()
            try:
                await self._read_task
            except asyncio.CancelledError:
                pass
            self._read_task = None
        if self._writer:
            try:
                self._writer.close()
                await self._writer.wait_closed()
            except Exception:
                pass
            self._writer = None
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(ConnectionError("disconnected"))
        self._pending.clear()

    async def _read_loop(self) -> None:
        assert self._reader is not None
        try:
            while True:
                frame = await protocol.read_frame(self._reader)
                if "push" in frame:
                    self._on_push(frame)
                else:
                    fut = self._pending.pop(frame.get("i"), None)
                    if fut and not fut.done():
                        if frame.get("ok"):
                            fut.set_result(frame)
                        else:
                            fut.set_exception(RuntimeError(frame.get("error", "broker error")))
        except (asyncio.IncompleteReadError, ConnectionError, asyncio.CancelledError):
            pass
        finally:
            if not self._closed:
                self._closed = True
                for fut in self._pending.values():
                    if not fut.done():
                        fut.set_exception(ConnectionError("broker connection lost"))
                self._pending.clear()

    def _on_push(self, frame: Dict[str, Any]) -> None:
        if frame.get("push") != "deliver":
            return
        delivery = Delivery(self, frame)
        cb = self._queue_callbacks.get(delivery.queue)
        if cb is not None:
            t = asyncio.ensure_future(self._run_cb(cb, delivery))
        else:
            # No consumer (e.g. cancelled): requeue.
            t = asyncio.ensure_future(delivery.nack(requeue=True))
        self._cb_tasks.add(t)
        t.add_done_callback(self._cb_tasks.discard)

    @staticmethod
    async def _run_cb(cb: DeliverCallback, delivery: Delivery) -> None:
        try:
            await cb(delivery)
        except Exception:
            logger.exception("consumer callback raised; requeueing %s", delivery.tag)
            await delivery.nack(requeue=True, error="callback crashed")

    # -- request/response ------------------------------------------------

    async def _send(self, frame: Dict[str, Any]) -> None:
        if self._writer is None or self._closed:
            raise ConnectionError("not connected")
        async with self._send_lock:
            self._writer.write(protocol.encode(frame))
            await self._writer.drain()

    async def call(self, frame: Dict[str, Any], timeout: float = 30.0) -> Dict[str, Any]:
        self._next_i += 1
        i = self._next_i
        frame = dict(frame, i=i)
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        self._pending[i] = fut
        await self._send(frame)
        return await asyncio.wait_for(fut, timeout)

    # -- queue infrastructure -------------------------------------------

    async def setup_queue_infrastructure(self, queue_name: str) -> None:
        ttl = self.config.job_ttl_ms or None
        await self.call({"m": "declare", "queue": queue_name, "durable": True, "ttl_ms": ttl})
        await self.call({"m": "declare", "queue": queue_name + RESULTS_SUFFIX, "durable": True})

    async def setup_pipeline_infrastructure(self, pipeline: PipelineConfig) -> None:
        for stage in pipeline.stages:
            await self.call(
                {"m": "declare", "queue": pipeline.get_stage_queue_name(stage.name), "durable": True}
            )
        await self.call(
            {"m": "declare", "queue": pipeline.get_pipeline_results_queue_name(), "durable": True}
        )

    # -- publish ---------------------------------------------------------

    async def publish_job(self, queue_name: str, job: Job) -> None:
        await self.call(
            {"m": "publish", "queue": queue_name, "body": job.model_dump_json(), "id": job.id}
        )

    async def publish_jobs(self, queue_name: str, jobs: List[Job]) -> None:
        items = [{"body": j.model_dump_json(), "id": j.id} for j in jobs]
        await self.call({"m": "publish_batch", "queue": queue_name, "items": items}, timeout=120.0)

    async def publish_result(self, queue_name: str, result: Result) -> None:
        target = (
            queue_name
            if queue_name.endswith(RESULTS_SUFFIX)
            else queue_name + RESULTS_SUFFIX
        )
        await self.call(
            {"m": "publish", "queue": target, "body": result.model_dump_json(), "id": result.id}
        )

    async def publish_to_queue(self, queue_name: str, body: str, msg_id: str = "") -> None:
        await self.call({"m": "publish", "queue": queue_name, "body": body, "id": msg_id})

    # -- consume ---------------------------------------------------------

    async def consume(
        self, queue_name: str, callback: DeliverCallback, prefetch: Optional[int] = None
    ) -> int:
        self._queue_callbacks[queue_name] = callback
        try:
            reply = await self.call(
                {
                    "m": "consume",
                    "queue": queue_name,
                    "prefetch": prefetch if prefetch is not None else self.default_prefetch,
                }
            )
        except Exception:
            self._queue_callbacks.pop(queue_name, None)
            raise
        cid = reply["consumer_id"]
        self._consumer_queue[cid] = queue_name
        return cid

    async def consume_jobs(
        self, queue_name: str, callback: DeliverCallback, prefetch: Optional[int] = None
    ) -> int:
        return await self.consume(queue_name, callback, prefetch)

    async def consume_results(
        self, queue_name: str, callback: DeliverCallback, prefetch: Optional[int] = None
    ) -> int:
        target = (
            queue_name
            if queue_name.endswith(RESULTS_SUFFIX)
            else queue_name + RESULTS_SUFFIX
        )
        return await self.consume(target, callback, prefetch)

    async def cancel_consumer(self, consumer_id: int) -> None:
        queue = self._consumer_queue.pop(consumer_id, None)
        if queue is not None:
            self._queue_callbacks.pop(queue, None)
        try:
            await self.call({"m": "cancel", "consumer_id": consumer_id})
        except (ConnectionError, RuntimeError):
            pass

    # -- stats / admin ---------------------------------------------------

    async def get_queue_stats(self, queue_name: str) -> QueueStats:
        try:
            reply = await self.call({"m": "stats", "queue": queue_name})
            return QueueStats(**reply["stats"], stats_source="broker")
        except RuntimeError:
            return QueueStats(queue_name=queue_name, stats_source="unavailable")
        except ConnectionError:
            return QueueStats(queue_name=queue_name, stats_source="unavailable")

    async def list_queues(self) -> List[QueueStats]:
        reply = await self.call({"m": "list"})
        return [QueueStats(**s, stats_source="broker") for s in reply["queues"]]

    async def get_failed_messages(self, queue_name: str, limit: int = 10) -> List[ErrorInfo]:
        target = queue_name if queue_name.endswith(FAILED_SUFFIX) else queue_name + FAILED_SUFFIX
        reply = await self.call({"m": "peek", "queue": target, "limit": limit})
        errors: List[ErrorInfo] = []
        for body in reply["messages"]:
            try:
                blob = json.loads(body)
                errors.append(
                    ErrorInfo(
                        job_id=blob.get("job_id", "unknown"),
                        error_message=blob.get("error", "unknown"),
                        timestamp=blob.get("timestamp", time.time()),
                        worker_id=blob.get("worker_id") or None,
                    )
                )
            except (json.JSONDecodeError, ValueError):
                errors.append(
                    ErrorInfo(job_id="unknown", error_message=body[:200], timestamp=time.time())
                )
        return errors

    async def clear_queue(self, queue_name: str) -> int:
        reply = await self.call({"m": "purge", "queue": queue_name})
        return reply.get("purged", 0)

    async def heartbeat(
        self,
        worker_id: str,
        queue: str,
        jobs_processed: int,
        avg_duration_ms: Optional[float],
        status: str = "active",
    ) -> None:
        await self.call(
            {
                "m": "heartbeat",
                "worker_id": worker_id,
                "queue": queue,
                "jobs_processed": jobs_processed,
                "avg_duration_ms": avg_duration_ms,
                "status": status,
            }
        )

    async def get_workers(self) -> List[Dict[str, Any]]:
        reply = await self.call({"m": "workers"})
        return reply["workers"]
