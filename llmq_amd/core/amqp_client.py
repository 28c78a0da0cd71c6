"""AMQP 0-9-1 BrokerClient backend.

Selected by URL scheme (``LLMQ_BROKER_URL=amqp://user:pass@host:port/vhost``
— the reference's RABBITMQ_URL form, /root/reference/llmq/core/config.py),
so an existing RabbitMQ deployment can back this framework exactly as it
backs the reference (BrokerManager over aio-pika, reference
core/broker.py:18-353). Speaks the protocol directly via
``broker/amqp_codec.py`` (no aio-pika in the image); the same client also
works against the in-tree broker's AMQP front-end, which is how it is
integration-tested offline. Set ``RABBITMQ_URL`` to run the test suite
against a real RabbitMQ (mirrors reference tests/test_integration.py:19-21).

Parity notes vs the JSON backend:
- publisher confirms (confirm.select) stand in for the RPC-acked publish;
- queue stats come from passive queue.declare (message_count ready +
  consumer_count — the reference's "amqp_fallback" source, broker.py:235);
- failed-job peek = basic.get + nack(requeue) on ``<q>.failed`` (the
  reference's get_failed_messages, broker.py:291-338);
- worker heartbeat/registry has no AMQP equivalent: heartbeat() is a no-op
  and get_workers() returns [] (health falls back to queue stats, like the
  reference).
"""

from __future__ import annotations

import asyncio
import logging
import time
import urllib.parse
from typing import Any, Dict, List, Optional, Tuple

from llmq_amd.broker import amqp_codec as c
from llmq_amd.broker.server import FAILED_SUFFIX, RESULTS_SUFFIX
from llmq_amd.core import client as _client_mod
from llmq_amd.core.client import BrokerClient, Delivery
from llmq_amd.core.config import Config
from llmq_amd.core.models import ErrorInfo, Job, QueueStats, Result
from llmq_amd.core.pipeline import PipelineConfig

logger = logging.getLogger(__name__)

CH = 1  # all traffic on one channel


class ChannelClosed(Exception):
    def __init__(self, code: int, text: str):
        super().__init__(f"channel closed: {code} {text}")
        self.code = code


class AMQPBrokerClient(BrokerClient):
    def __init__(self, config: Optional[Config] = None):
        super().__init__(config)
        u = urllib.parse.urlsplit(self.config.broker_url)
        self.amqp_host = u.hostname or "127.0.0.1"
        self.amqp_port = u.port or 5672
        self.amqp_user = u.username or "guest"
        self.amqp_password = u.password or "guest"
        vhost = u.path[1:] if u.path.startswith("/") else u.path
        self.amqp_vhost = urllib.parse.unquote(vhost) or "/"
        self._rpc_lock = asyncio.Lock()
        self._rpc_expect: Tuple[str, ...] = ()
        self._rpc_future: Optional[asyncio.Future] = None
        self._chan_open = False
        # publisher confirms
        self._pub_seq = 0
        self._confirmed = 0
        self._confirm_waiters: List[Tuple[int, asyncio.Future]] = []
        # content assembly: (kind, args) + props/body while assembling
        self._asm: Optional[Dict[str, Any]] = None
        self._ctag_queue: Dict[str, str] = {}
        self._cid_ctag: Dict[int, str] = {}
        self._next_cid = 0
        self._get_future: Optional[asyncio.Future] = None
        self._mgmt_ok: Optional[bool] = None  # memoised mgmt-API reachability

    # -- connection ------------------------------------------------------

    async def connect(self, retries: int = 5) -> None:
        delay = 0.5
        last_exc: Optional[Exception] = None
        for attempt in range(retries):
            try:
                await self._connect_once()
                return
            except (ConnectionError, OSError, asyncio.TimeoutError, c.AMQPError) as exc:
                last_exc = exc
                logger.warning(
                    "amqp connect attempt %d/%d failed: %s", attempt + 1, retries, exc
                )
                await asyncio.sleep(delay)
                delay = min(delay * 2, 8.0)
        raise ConnectionError(
            f"Could not connect to AMQP broker at {self.config.broker_url}: {last_exc}"
        )

    async def _connect_once(self) -> None:
        # fresh connection = fresh channel: confirm tags restart server-side
        self._pub_seq = 0
        self._confirmed = 0
        for _seq, f in self._confirm_waiters:
            if not f.done():
                f.set_exception(ConnectionError("reconnected"))
        self._confirm_waiters.clear()
        self._chan_open = False
        self._asm = None
        self._reader, self._writer = await asyncio.open_connection(
            self.amqp_host, self.amqp_port
        )
        self._writer.write(c.PROTOCOL_HEADER)
        await self._writer.drain()
        name, _ = await self._read_method_handshake("connection.start")
        self._writer.write(c.method_frame(
            0, "connection.start-ok",
            client_properties={
                "product": "llmq-amd", "version": "2.0",
                "capabilities": {"basic.nack": True, "publisher_confirms": True},
            },
            mechanism="PLAIN",
            response=b"\x00" + self.amqp_user.encode() + b"\x00" + self.amqp_password.encode(),
            locale="en_US",
        ))
        name, args = await self._read_method_handshake("connection.tune")
        frame_max = int(args.get("frame_max") or c.DEFAULT_FRAME_MAX)
        self._frame_max = frame_max if frame_max else c.DEFAULT_FRAME_MAX
        # heartbeat 0 disables the timer on both sides (tolerated by RabbitMQ)
        self._writer.write(c.method_frame(
            0, "connection.tune-ok", channel_max=int(args.get("channel_max") or 0),
            frame_max=self._frame_max, heartbeat=0,
        ))
        self._writer.write(c.method_frame(
            0, "connection.open", virtual_host=self.amqp_vhost, reserved1="",
            reserved2=False,
        ))
        await self._writer.drain()
        await self._read_method_handshake("connection.open-ok")
        self._closed = False
        self._read_task = asyncio.create_task(self._read_loop())
        await self._open_channel()

    async def _read_method_handshake(self, expected: str) -> Tuple[str, Dict[str, Any]]:
        while True:
            ftype, _ch, payload = await asyncio.wait_for(c.read_frame(self._reader), 30)
            if ftype == c.FRAME_HEARTBEAT:
                continue
            name, args = c.decode_method(payload)
            if name != expected:
                raise c.AMQPError(f"handshake expected {expected}, got {name}")
            return name, args

    async def _open_channel(self) -> None:
        await self._rpc("channel.open", ("channel.open-ok",), reserved1="")
        self._chan_open = True
        await self._rpc("confirm.select", ("confirm.select-ok",))

    async def disconnect(self) -> None:
        if self._writer is not None and not self._closed:
            try:
                self._writer.write(c.method_frame(
                    0, "connection.close", reply_code=200, reply_text="bye",
                    class_id=0, method_id=0,
                ))
                await self._writer.drain()
            except Exception:
                pass
        await super().disconnect()

    # -- frame pump ------------------------------------------------------

    async def _read_loop(self) -> None:
        try:
            while True:
                ftype, ch, payload = await c.read_frame(self._reader)
                if ftype == c.FRAME_HEARTBEAT:
                    continue
                if ftype == c.FRAME_METHOD:
                    self._on_method(ch, *c.decode_method(payload))
                elif ftype == c.FRAME_HEADER:
                    if self._asm is not None:
                        size, props = c.decode_content_header(payload)
                        self._asm["size"] = size
                        self._asm["props"] = props
                        self._asm["body"] = b""
                        if size == 0:
                            self._finish_content()
                elif ftype == c.FRAME_BODY:
                    if self._asm is not None:
                        self._asm["body"] += payload
                        if len(self._asm["body"]) >= self._asm["size"]:
                            self._finish_content()
        except (asyncio.IncompleteReadError, ConnectionError, asyncio.CancelledError,
                c.AMQPError):
            pass
        finally:
            if not self._closed:
                self._closed = True
            fut = self._rpc_future
            if fut is not None and not fut.done():
                fut.set_exception(ConnectionError("amqp connection lost"))
            if self._get_future is not None and not self._get_future.done():
                self._get_future.set_exception(ConnectionError("amqp connection lost"))
            for _seq, f in self._confirm_waiters:
                if not f.done():
                    f.set_exception(ConnectionError("amqp connection lost"))
            self._confirm_waiters.clear()

    def _on_method(self, ch: int, name: str, args: Dict[str, Any]) -> None:
        if name == "basic.deliver":
            self._asm = {"kind": "deliver", "args": args}
            return
        if name == "basic.get-ok":
            self._asm = {"kind": "get", "args": args}
            return
        if name == "basic.get-empty":
            if self._get_future is not None and not self._get_future.done():
                self._get_future.set_result(None)
            return
        if name == "basic.ack" and ch == CH:  # publisher confirm
            tag = int(args["delivery_tag"])
            self._confirmed = max(self._confirmed, tag) if args.get("multiple") else \
                max(self._confirmed, tag)
            still = []
            for seq, f in self._confirm_waiters:
                if seq <= self._confirmed:
                    if not f.done():
                        f.set_result(True)
                else:
                    still.append((seq, f))
            self._confirm_waiters = still
            return
        if name == "channel.close":
            self._chan_open = False
            exc = ChannelClosed(args.get("reply_code", 0), args.get("reply_text", ""))
            try:
                self._writer.write(c.method_frame(ch, "channel.close-ok"))
            except Exception:
                pass
            fut = self._rpc_future
            if fut is not None and not fut.done():
                fut.set_exception(exc)
            if self._get_future is not None and not self._get_future.done():
                self._get_future.set_exception(exc)
            return
        if name == "connection.close":
            try:
                self._writer.write(c.method_frame(0, "connection.close-ok"))
            except Exception:
                pass
            self._closed = True
            return
        if name == "basic.cancel":  # consumer-cancel notify
            self._ctag_queue.pop(args.get("consumer_tag", ""), None)
            return
        fut = self._rpc_future
        if fut is not None and not fut.done() and name in self._rpc_expect:
            fut.set_result((name, args))
            return
        logger.debug("unexpected amqp method %s on ch %d", name, ch)

    def _finish_content(self) -> None:
        asm, self._asm = self._asm, None
        args = asm["args"]
        body = asm["body"].decode("utf-8", "replace")
        if asm["kind"] == "get":
            if self._get_future is not None and not self._get_future.done():
                self._get_future.set_result((args, asm.get("props", {}), body))
            return
        queue = self._ctag_queue.get(args.get("consumer_tag", ""),
                                     args.get("routing_key", ""))
        headers = (asm.get("props") or {}).get("headers") or {}
        frame = {
            "queue": queue,
            "tag": args["delivery_tag"],
            "body": body,
            "redelivered": bool(args.get("redelivered")),
            "attempts": int(headers.get("x-attempts") or
                            (2 if args.get("redelivered") else 1)),
        }
        delivery = Delivery(self, frame)
        cb = self._queue_callbacks.get(queue)
        if cb is not None:
            t = asyncio.ensure_future(self._run_cb(cb, delivery))
        else:
            t = asyncio.ensure_future(delivery.nack(requeue=True))
        self._cb_tasks.add(t)
        t.add_done_callback(self._cb_tasks.discard)

    # -- plumbing --------------------------------------------------------

    async def _write(self, data: bytes) -> None:
        if self._writer is None or self._closed:
            raise ConnectionError("not connected")
        async with self._send_lock:
            self._writer.write(data)
            await self._writer.drain()

    async def _rpc(self, name: str, expect: Tuple[str, ...], timeout: float = 30.0,
                   **args: Any) -> Dict[str, Any]:
        async with self._rpc_lock:
            self._rpc_expect = expect
            fut: asyncio.Future = asyncio.get_event_loop().create_future()
            self._rpc_future = fut
            await self._write(c.method_frame(CH, name, **args))
            try:
                _name, reply = await asyncio.wait_for(fut, timeout)
            finally:
                self._rpc_future = None
                self._rpc_expect = ()
            return reply

    async def _ensure_channel(self) -> None:
        if not self._chan_open:
            await self._open_channel()

    # Delivery.ack()/nack() route through _send with the JSON-client dicts.
    async def _send(self, frame: Dict[str, Any]) -> None:
        m = frame.get("m")
        if m == "ack":
            await self._write(c.method_frame(
                CH, "basic.ack", delivery_tag=frame["tag"], multiple=False))
        elif m == "nack":
            await self._write(c.method_frame(
                CH, "basic.nack", delivery_tag=frame["tag"], multiple=False,
                requeue=bool(frame.get("requeue", True))))
        else:
            raise RuntimeError(f"AMQP backend cannot send raw frame {m}")

    # -- queue infrastructure -------------------------------------------

    async def _declare(self, queue: str, ttl_ms: Optional[int] = None,
                       dlq_for: Optional[str] = None) -> Dict[str, Any]:
        arguments: Dict[str, Any] = {}
        if ttl_ms:
            arguments["x-message-ttl"] = int(ttl_ms)
        if dlq_for:
            # RabbitMQ: rejects / TTL expiries route to <q>.failed via the
            # default exchange (the real DLQ the reference never wired up)
            arguments["x-dead-letter-exchange"] = ""
            arguments["x-dead-letter-routing-key"] = dlq_for
        await self._ensure_channel()
        return await self._rpc(
            "queue.declare", ("queue.declare-ok",), queue=queue, passive=False,
            durable=True, exclusive=False, auto_delete=False, nowait=False,
            arguments=arguments,
        )

    async def setup_queue_infrastructure(self, queue_name: str) -> None:
        await self._declare(queue_name + FAILED_SUFFIX)
        await self._declare(queue_name, ttl_ms=self.config.job_ttl_ms or None,
                            dlq_for=queue_name + FAILED_SUFFIX)
        await self._declare(queue_name + RESULTS_SUFFIX)

    async def setup_pipeline_infrastructure(self, pipeline: PipelineConfig) -> None:
        for stage in pipeline.stages:
            qn = pipeline.get_stage_queue_name(stage.name)
            await self._declare(qn + FAILED_SUFFIX)
            await self._declare(qn, dlq_for=qn + FAILED_SUFFIX)
        await self._declare(pipeline.get_pipeline_results_queue_name())

    # -- publish ---------------------------------------------------------

    async def _publish_raw(self, queue: str, body: str, msg_id: str) -> int:
        props = {"delivery_mode": 2, "content_type": "application/json"}
        if msg_id:
            props["message_id"] = msg_id
        data = c.method_frame(CH, "basic.publish", reserved1=0, exchange="",
                              routing_key=queue, mandatory=False, immediate=False)
        data += c.content_frames(CH, body.encode("utf-8"), props,
                                 getattr(self, "_frame_max", c.DEFAULT_FRAME_MAX))
        await self._write(data)
        self._pub_seq += 1
        return self._pub_seq

    async def _wait_confirm(self, seq: int, timeout: float = 60.0) -> None:
        if self._confirmed >= seq:
            return
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        self._confirm_waiters.append((seq, fut))
        await asyncio.wait_for(fut, timeout)

    async def publish_job(self, queue_name: str, job: Job) -> None:
        seq = await self._publish_raw(queue_name, job.model_dump_json(), job.id)
        await self._wait_confirm(seq)

    async def publish_jobs(self, queue_name: str, jobs: List[Job]) -> None:
        seq = 0
        for j in jobs:
            seq = await self._publish_raw(queue_name, j.model_dump_json(), j.id)
        if seq:
            await self._wait_confirm(seq, timeout=120.0)

    async def publish_result(self, queue_name: str, result: Result) -> None:
        target = (queue_name if queue_name.endswith(RESULTS_SUFFIX)
                  else queue_name + RESULTS_SUFFIX)
        seq = await self._publish_raw(target, result.model_dump_json(), result.id)
        await self._wait_confirm(seq)

    async def publish_to_queue(self, queue_name: str, body: str, msg_id: str = "") -> None:
        seq = await self._publish_raw(queue_name, body, msg_id)
        await self._wait_confirm(seq)

    # -- consume ---------------------------------------------------------

    async def consume(self, queue_name: str, callback, prefetch: Optional[int] = None) -> int:
        await self._ensure_channel()
        self._queue_callbacks[queue_name] = callback
        try:
            await self._rpc(
                "basic.qos", ("basic.qos-ok",), prefetch_size=0,
                prefetch_count=min(65535, prefetch if prefetch is not None
                                   else self.default_prefetch),
                **{"global": False},
            )
            reply = await self._rpc(
                "basic.consume", ("basic.consume-ok",), reserved1=0,
                queue=queue_name, consumer_tag="", no_local=False, no_ack=False,
                exclusive=False, nowait=False, arguments={},
            )
        except Exception:
            self._queue_callbacks.pop(queue_name, None)
            raise
        ctag = reply["consumer_tag"]
        self._ctag_queue[ctag] = queue_name
        self._next_cid += 1
        cid = self._next_cid
        self._cid_ctag[cid] = ctag
        self._consumer_queue[cid] = queue_name
        return cid

    async def cancel_consumer(self, consumer_id: int) -> None:
        queue = self._consumer_queue.pop(consumer_id, None)
        if queue is not None:
            self._queue_callbacks.pop(queue, None)
        ctag = self._cid_ctag.pop(consumer_id, None)
        if ctag is None:
            return
        self._ctag_queue.pop(ctag, None)
        try:
            await self._rpc("basic.cancel", ("basic.cancel-ok",),
                            consumer_tag=ctag, nowait=False)
        except (ConnectionError, RuntimeError, ChannelClosed, asyncio.TimeoutError):
            pass

    # -- stats / admin ---------------------------------------------------

    # RabbitMQ management HTTP API (reference broker.py:244-289): full
    # stats incl. message bytes + queue listing. Port = AMQP port + 10000
    # (RabbitMQ's 5672 -> 15672 convention) unless LLMQ_MGMT_URL overrides.
    def _mgmt_base(self) -> str:
        import os

        return os.environ.get(
            "LLMQ_MGMT_URL",
            f"http://{self.amqp_host}:{self.amqp_port + 10000}",
        ).rstrip("/")

    async def _mgmt_get(self, path: str):
        import httpx

        vhost = urllib.parse.quote(self.amqp_vhost, safe="")
        async with httpx.AsyncClient(timeout=5.0) as client:
            r = await client.get(
                f"{self._mgmt_base()}{path.format(vhost=vhost)}",
                auth=(self.amqp_user, self.amqp_password),
            )
            r.raise_for_status()
            return r.json()

    @staticmethod
    def _stats_from_mgmt(blob: Dict[str, Any]) -> QueueStats:
        return QueueStats(
            queue_name=blob.get("name", ""),
            message_count=blob.get("messages", 0) or 0,
            message_count_ready=blob.get("messages_ready", 0) or 0,
            message_count_unacknowledged=blob.get("messages_unacknowledged", 0) or 0,
            consumer_count=blob.get("consumers", 0) or 0,
            message_bytes=blob.get("message_bytes", 0) or 0,
            message_bytes_ready=blob.get("message_bytes_ready", 0) or 0,
            message_bytes_unacknowledged=blob.get(
                "message_bytes_unacknowledged", 0) or 0,
            stats_source="management_api",
        )

    async def get_queue_stats(self, queue_name: str) -> QueueStats:
        # 1) management API (full stats); 2) passive declare fallback
        # (the reference's amqp_fallback, broker.py:235-239)
        if self._mgmt_ok is not False:
            try:
                blob = await self._mgmt_get(
                    "/api/queues/{vhost}/" + urllib.parse.quote(queue_name, safe=""))
                self._mgmt_ok = True
                return self._stats_from_mgmt(blob)
            except Exception as exc:  # noqa: BLE001 — mgmt API optional
                import httpx

                if isinstance(exc, httpx.HTTPStatusError) and exc.response.status_code == 404:
                    self._mgmt_ok = True  # API reachable; queue absent
                    return QueueStats(queue_name=queue_name,
                                      stats_source="unavailable")
                self._mgmt_ok = False
        try:
            await self._ensure_channel()
            reply = await self._rpc(
                "queue.declare", ("queue.declare-ok",), queue=queue_name,
                passive=True, durable=False, exclusive=False, auto_delete=False,
                nowait=False, arguments={},
            )
            return QueueStats(
                queue_name=queue_name,
                message_count=reply["message_count"],
                message_count_ready=reply["message_count"],
                consumer_count=reply["consumer_count"],
                stats_source="amqp_fallback",
            )
        except (ChannelClosed, ConnectionError, asyncio.TimeoutError):
            return QueueStats(queue_name=queue_name, stats_source="unavailable")

    async def list_queues(self) -> List[QueueStats]:
        try:
            blobs = await self._mgmt_get("/api/queues/{vhost}")
            return [self._stats_from_mgmt(b) for b in blobs]
        except Exception as exc:  # noqa: BLE001
            raise RuntimeError(
                "queue listing over AMQP needs the RabbitMQ management API "
                f"({self._mgmt_base()}; set LLMQ_MGMT_URL) — or use the "
                "in-tree broker (llmq:// URL)"
            ) from exc

    async def _basic_get(self, queue: str):
        async with self._rpc_lock:
            self._get_future = asyncio.get_event_loop().create_future()
            try:
                await self._write(c.method_frame(
                    CH, "basic.get", reserved1=0, queue=queue, no_ack=False))
                return await asyncio.wait_for(self._get_future, 30)
            finally:
                self._get_future = None

    async def get_failed_messages(self, queue_name: str, limit: int = 10) -> List[ErrorInfo]:
        import json as _json

        target = (queue_name if queue_name.endswith(FAILED_SUFFIX)
                  else queue_name + FAILED_SUFFIX)
        await self._ensure_channel()
        out: List[ErrorInfo] = []
        tags: List[int] = []
        try:
            for _ in range(limit):
                got = await self._basic_get(target)
                if got is None:
                    break
                args, props, body = got
                tags.append(args["delivery_tag"])
                try:
                    blob = _json.loads(body)
                    if isinstance(blob, dict) and "error" in blob:
                        out.append(ErrorInfo(
                            job_id=blob.get("job_id", props.get("message_id", "unknown")),
                            error_message=blob.get("error", "unknown"),
                            timestamp=blob.get("timestamp", time.time()),
                            worker_id=blob.get("worker_id") or None,
                        ))
                    else:  # a raw dead-lettered job (RabbitMQ DLX path)
                        out.append(ErrorInfo(
                            job_id=blob.get("id", "unknown"),
                            error_message="dead-lettered",
                            timestamp=time.time(),
                        ))
                except (ValueError, AttributeError):
                    out.append(ErrorInfo(job_id="unknown", error_message=body[:200],
                                         timestamp=time.time()))
        except ChannelClosed:
            return out
        # peek semantics: put them back
        for tag in tags:
            await self._write(c.method_frame(
                CH, "basic.nack", delivery_tag=tag, multiple=False, requeue=True))
        return out

    async def clear_queue(self, queue_name: str) -> int:
        await self._ensure_channel()
        try:
            reply = await self._rpc("queue.purge", ("queue.purge-ok",),
                                    reserved1=0, queue=queue_name, nowait=False)
            return int(reply["message_count"])
        except ChannelClosed:
            return 0

    async def heartbeat(self, worker_id: str, queue: str, jobs_processed: int,
                        avg_duration_ms: Optional[float], status: str = "active") -> None:
        return None  # no worker registry over plain AMQP (reference parity)

    async def get_workers(self) -> List[Dict[str, Any]]:
        return []


# register with the factory in core.client
_client_mod.AMQP_CLIENT_CLS = AMQPBrokerClient
