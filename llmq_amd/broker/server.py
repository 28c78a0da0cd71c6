"""The in-tree message broker.

One asyncio process owning all queues — the coordination backbone of the
framework, replacing the reference's external RabbitMQ (SURVEY §1 layer 0;
reference launches it via docker/singularity, README.md:54-59).

Semantics (matching what the reference relies on, plus its documented gaps
fixed — SURVEY §2 quirks):
- durable queues: persistent spool via ``Journal``; submitted jobs and
  unconsumed results survive broker restarts (resumable receive).
- per-consumer prefetch: at most ``prefetch`` unacked deliveries in flight
  per consumer; round-robin across consumers of a queue (data parallelism
  across N worker processes on one queue).
- at-least-once: ack retires; connection loss or nack(requeue) redelivers.
- real dead-letter queue: attempts are counted; a job nacked (or
  redelivered) more than ``max_retries`` times lands in ``<q>.failed`` with
  error info instead of looping forever (the reference requeues poison jobs
  forever, base.py:245).
- TTL: queues may declare a message TTL; expired messages dead-letter
  (the reference's job_ttl_ms exists but is never applied, config.py:46-64).
- worker health registry: workers heartbeat; ``llmq health`` reads real
  per-worker status (the reference infers health from queue stats only).
"""

from __future__ import annotations

import asyncio
import logging
import os
import time
from collections import deque
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Dict, List, Optional, Tuple

from llmq_amd.broker import protocol
from llmq_amd.broker.journal import Journal

logger = logging.getLogger(__name__)

FAILED_SUFFIX = ".failed"
RESULTS_SUFFIX = ".results"


@dataclass
class Message:
    seq: int
    msg_id: str
    body: str
    attempts: int = 0
    enqueued_at: float = field(default_factory=time.time)

    @property
    def nbytes(self) -> int:
        return len(self.body)


@dataclass
class Consumer:
    conn: "Connection"
    consumer_id: int
    queue: str
    prefetch: int
    inflight: int = 0
    cancelled: bool = False


class Queue:
    def __init__(
        self,
        name: str,
        durable: bool,
        data_dir: Optional[Path],
        ttl_ms: int = 0,
        max_retries: int = 3,
        journal_fsync: bool = False,
    ):
        self.name = name
        self.durable = durable
        self.ttl_ms = ttl_ms
        self.max_retries = max_retries
        self.ready: deque[Message] = deque()
        self.unacked: Dict[Tuple[int, int], Message] = {}  # (conn_id, tag) -> msg
        self.consumers: List[Consumer] = []
        self._rr = 0
        self._seq = 0
        self.journal: Optional[Journal] = None
        if durable and data_dir is not None:
            self.journal = Journal(data_dir, name, fsync=journal_fsync)
            for seq, msg_id, body, attempts in self.journal.load():
                self.ready.append(Message(seq, msg_id, body, attempts))
                self._seq = max(self._seq, seq)

    def next_seq(self) -> int:
        self._seq += 1
        return self._seq

    # -- stats -----------------------------------------------------------

    def stats(self) -> Dict[str, Any]:
        ready_bytes = sum(m.nbytes for m in self.ready)
        unacked_bytes = sum(m.nbytes for m in self.unacked.values())
        return {
            "queue_name": self.name,
            "message_count": len(self.ready) + len(self.unacked),
            "message_count_ready": len(self.ready),
            "message_count_unacknowledged": len(self.unacked),
            "consumer_count": sum(1 for c in self.consumers if not c.cancelled),
            "message_bytes": ready_bytes + unacked_bytes,
            "message_bytes_ready": ready_bytes,
            "message_bytes_unacknowledged": unacked_bytes,
        }

    def live_messages(self):
        for m in self.ready:
            yield (m.seq, m.msg_id, m.body, m.attempts)
        for m in self.unacked.values():
            yield (m.seq, m.msg_id, m.body, m.attempts)


class Connection:
    _next_id = 0

    def __init__(self, server: "BrokerServer", writer: asyncio.StreamWriter):
        Connection._next_id += 1
        self.id = Connection._next_id
        self.server = server
        self.writer = writer
        self.consumers: Dict[int, Consumer] = {}
        self._next_tag = 0
        self._next_consumer = 0
        self.tags: Dict[int, Tuple[Queue, Consumer]] = {}  # tag -> owner
        self._send_lock = asyncio.Lock()
        self.closed = False

    def next_tag(self) -> int:
        self._next_tag += 1
        return self._next_tag

    def next_consumer_id(self) -> int:
        self._next_consumer += 1
        return self._next_consumer

    async def send(self, frame: Dict[str, Any]) -> None:
        if self.closed:
            return
        async with self._send_lock:
            try:
                self.writer.write(protocol.encode(frame))
                await self.writer.drain()
            except (ConnectionError, RuntimeError):
                self.closed = True


class BrokerServer:
    """asyncio broker. ``await serve()`` binds; ``close()`` shuts down."""

    def __init__(
        self,
        host: str = "127.0.0.1",
        port: int = 5672,
        data_dir: Optional[str] = None,
        max_retries: int = 3,
        default_ttl_ms: int = 0,
        journal_fsync: Optional[bool] = None,
    ):
        self.host = host
        self.port = port
        self.data_dir = Path(data_dir) if data_dir else None
        # Durability window: by default journals are flush()ed (no fsync)
        # every 50 ms — a crash of the HOST (not just the broker process)
        # can lose the last <=50 ms of publishes. LLMQ_JOURNAL_FSYNC=1 (or
        # journal_fsync=True) fsyncs at every publish(-batch) boundary
        # BEFORE the client's confirm, matching RabbitMQ's
        # persistent-message guarantee at a throughput cost.
        if journal_fsync is None:
            journal_fsync = os.environ.get("LLMQ_JOURNAL_FSYNC", "").lower() in (
                "1", "true", "yes", "on")
        self.journal_fsync = journal_fsync
        self.max_retries = max_retries
        self.default_ttl_ms = default_ttl_ms
        self.queues: Dict[str, Queue] = {}
        self.workers: Dict[str, Dict[str, Any]] = {}  # worker_id -> health blob
        self._send_tasks: set = set()  # strong refs: bare ensure_future may be GC'd
        self._server: Optional[asyncio.AbstractServer] = None
        self._dirty_journals: set[str] = set()
        self._flusher_task: Optional[asyncio.Task] = None
        self.started_at = time.time()

    # -- lifecycle -------------------------------------------------------

    async def serve(self) -> None:
        if self.data_dir is not None:
            self.data_dir.mkdir(parents=True, exist_ok=True)
            # Recover every spooled queue at startup.
            for path in sorted(self.data_dir.glob("*.jsonl")):
                name = self._unescape(path.stem)
                if name not in self.queues:
                    self._declare(name, durable=True)
        self._server = await asyncio.start_server(
            self._handle_conn, self.host, self.port, limit=protocol.MAX_FRAME
        )
        self._flusher_task = asyncio.create_task(self._flush_loop())
        addr = self._server.sockets[0].getsockname()
        self.port = addr[1]
        logger.info("broker listening on %s:%d", addr[0], addr[1])

    @staticmethod
    def _unescape(stem: str) -> str:
        out, i = [], 0
        while i < len(stem):
            if stem[i] == "%" and i + 2 < len(stem) + 1 and i + 3 <= len(stem):
                try:
                    out.append(chr(int(stem[i + 1 : i + 3], 16)))
                    i += 3
                    continue
                except ValueError:
                    pass
            out.append(stem[i])
            i += 1
        return "".join(out)

    async def close(self) -> None:
        if self._flusher_task:
            self._flusher_task.cancel()
            try:
                await self._flusher_task
            except asyncio.CancelledError:
                pass
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        for q in self.queues.values():
            if q.journal:
                q.journal.close()

    async def _flush_loop(self) -> None:
        while True:
            await asyncio.sleep(0.05)
            self._flush_dirty()

    def _flush_dirty(self) -> None:
        for name in list(self._dirty_journals):
            q = self.queues.get(name)
            if q and q.journal:
                q.journal.flush()
                q.journal.maybe_compact(q.live_messages())
        self._dirty_journals.clear()

    # -- queue ops -------------------------------------------------------

    def _declare(self, name: str, durable: bool = True, ttl_ms: Optional[int] = None) -> Queue:
        q = self.queues.get(name)
        if q is None:
            q = Queue(
                name,
                durable,
                self.data_dir,
                ttl_ms if ttl_ms is not None else self.default_ttl_ms,
                self.max_retries,
                journal_fsync=self.journal_fsync,
            )
            self.queues[name] = q
        elif ttl_ms is not None:
            q.ttl_ms = ttl_ms
        return q

    def _publish(self, q: Queue, body: str, msg_id: str, attempts: int = 0) -> None:
        msg = Message(q.next_seq(), msg_id, body, attempts)
        q.ready.append(msg)
        if q.journal:
            q.journal.append_publish(msg.seq, msg.msg_id, msg.body, msg.attempts)
            self._dirty_journals.add(q.name)

    def _dead_letter(self, q: Queue, msg: Message, error: str, worker_id: str = "") -> None:
        import json as _json

        dlq = self._declare(q.name + FAILED_SUFFIX, durable=q.durable)
        blob = _json.dumps(
            {
                "job": msg.body,
                "job_id": msg.msg_id,
                "error": error,
                "worker_id": worker_id,
                "timestamp": time.time(),
                "attempts": msg.attempts,
                "queue": q.name,
            },
            separators=(",", ":"),
        )
        self._publish(dlq, blob, msg.msg_id)
        self._kick(dlq)

    def _kick(self, q: Queue) -> None:
        """Dispatch ready messages to consumers with free prefetch windows."""
        if not q.consumers:
            return
        now = time.time()
        while q.ready:
            active = [c for c in q.consumers if not c.cancelled and not c.conn.closed]
            if not active:
                return
            target = None
            for off in range(len(active)):
                c = active[(q._rr + off) % len(active)]
                if c.inflight < c.prefetch:
                    target = c
                    q._rr = (q._rr + off + 1) % len(active)
                    break
            if target is None:
                return
            msg = q.ready.popleft()
            if q.ttl_ms and (now - msg.enqueued_at) * 1000 > q.ttl_ms:
                self._ack_internal(q, msg)
                self._dead_letter(q, msg, "expired: TTL exceeded")
                continue
            if msg.attempts > q.max_retries:
                # poison job that kills its consumer: disconnect-requeued
                # deliveries count too, so the retry cap holds even when no
                # explicit nack ever arrives
                self._ack_internal(q, msg)
                self._dead_letter(
                    q, msg, f"max retries exceeded ({q.max_retries}): "
                    "consumer lost repeatedly",
                )
                continue
            tag = target.conn.next_tag()
            msg.attempts += 1
            q.unacked[(target.conn.id, tag)] = msg
            target.inflight += 1
            target.conn.tags[tag] = (q, target)
            t = asyncio.ensure_future(
                target.conn.send(
                    {
                        "push": "deliver",
                        "queue": q.name,
                        "tag": tag,
                        "body": msg.body,
                        "redelivered": msg.attempts > 1,
                        "attempts": msg.attempts,
                    }
                )
            )
            self._send_tasks.add(t)
            t.add_done_callback(self._send_tasks.discard)

    def _sync_if_required(self, q: Queue) -> None:
        """fsync-before-confirm durability (LLMQ_JOURNAL_FSYNC)."""
        if self.journal_fsync and q.journal:
            q.journal.flush()
            self._dirty_journals.discard(q.name)

    def _ack_internal(self, q: Queue, msg: Message) -> None:
        if q.journal:
            q.journal.append_ack(msg.seq)
            self._dirty_journals.add(q.name)

    # -- connection handling ---------------------------------------------

    async def _handle_conn(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        # Dual protocol on one port: an AMQP 0-9-1 client opens with the
        # 8-byte protocol header; the in-tree protocol's frames are
        # '{'-prefixed JSON lines. Sniff the first bytes.
        try:
            first = await reader.readexactly(1)
        except (asyncio.IncompleteReadError, ConnectionError):
            writer.close()
            return
        if first == b"A":
            try:
                rest = await reader.readexactly(7)
            except (asyncio.IncompleteReadError, ConnectionError):
                writer.close()
                return
            if rest[:3] == b"MQP":
                from llmq_amd.broker.amqp_server import handle_amqp_connection

                try:
                    await handle_amqp_connection(self, reader, writer)
                finally:
                    try:
                        writer.close()
                    except Exception:
                        pass
                return
            writer.close()
            return
        conn = Connection(self, writer)
        pending_first: Optional[bytes] = first
        try:
            while True:
                try:
                    if pending_first is not None:
                        line = pending_first + await reader.readline()
                        pending_first = None
                        if not line.endswith(b"\n"):
                            break
                        import json as _json

                        frame = _json.loads(line)
                    else:
                        frame = await protocol.read_frame(reader)
                except (asyncio.IncompleteReadError, ConnectionError):
                    break
                except ValueError:
                    break
                await self._handle_frame(conn, frame)
        finally:
            conn.closed = True
            self._on_disconnect(conn)
            try:
                writer.close()
            except Exception:
                pass

    def _on_disconnect(self, conn: Connection) -> None:
        # Requeue everything unacked by this connection (at-least-once).
        requeued = set()
        for tag, (q, consumer) in list(conn.tags.items()):
            msg = q.unacked.pop((conn.id, tag), None)
            if msg is not None:
                q.ready.appendleft(msg)
                requeued.add(q.name)
        for c in conn.consumers.values():
            c.cancelled = True
            q = self.queues.get(c.queue)
            if q and c in q.consumers:
                q.consumers.remove(c)
        for name in requeued:
            self._kick(self.queues[name])

    async def _handle_frame(self, conn: Connection, frame: Dict[str, Any]) -> None:
        i = frame.get("i")
        method = frame.get("m")
        try:
            result = await self._dispatch(conn, method, frame)
            if i is not None:
                reply = {"i": i, "ok": True}
                if result:
                    reply.update(result)
                await conn.send(reply)
        except Exception as exc:  # noqa: BLE001 — reported to the client
            logger.debug("method %s failed: %s", method, exc)
            if i is not None:
                await conn.send({"i": i, "ok": False, "error": str(exc)})

    async def _dispatch(self, conn, method, f) -> Optional[Dict[str, Any]]:
        if method == "ping":
            return {"pong": True, "uptime": time.time() - self.started_at}

        if method == "declare":
            q = self._declare(f["queue"], f.get("durable", True), f.get("ttl_ms"))
            self._kick(q)
            return {"queue": q.name}

        if method == "publish":
            q = self._declare(f["queue"])
            self._publish(q, f["body"], f.get("id", ""))
            self._sync_if_required(q)
            self._kick(q)
            return None

        if method == "publish_batch":
            q = self._declare(f["queue"])
            for item in f["items"]:
                self._publish(q, item["body"], item.get("id", ""))
            self._sync_if_required(q)
            self._kick(q)
            return {"count": len(f["items"])}

        if method == "consume":
            q = self._declare(f["queue"])
            c = Consumer(conn, conn.next_consumer_id(), q.name, max(1, int(f.get("prefetch", 1))))
            conn.consumers[c.consumer_id] = c
            q.consumers.append(c)
            self._kick(q)
            return {"consumer_id": c.consumer_id}

        if method == "cancel":
            c = conn.consumers.pop(int(f["consumer_id"]), None)
            if c:
                c.cancelled = True
                q = self.queues.get(c.queue)
                if q and c in q.consumers:
                    q.consumers.remove(c)
            return None

        if method == "ack":
            tag = int(f["tag"])
            owner = conn.tags.pop(tag, None)
            if owner is None:
                return None
            q, consumer = owner
            msg = q.unacked.pop((conn.id, tag), None)
            if msg is not None:
                self._ack_internal(q, msg)
            consumer.inflight = max(0, consumer.inflight - 1)
            self._kick(q)
            return None

        if method == "nack":
            tag = int(f["tag"])
            owner = conn.tags.pop(tag, None)
            if owner is None:
                return None
            q, consumer = owner
            msg = q.unacked.pop((conn.id, tag), None)
            consumer.inflight = max(0, consumer.inflight - 1)
            if msg is not None:
                error = f.get("error", "")
                if not f.get("requeue", True):
                    self._ack_internal(q, msg)
                    self._dead_letter(q, msg, error or "rejected", f.get("worker", ""))
                elif msg.attempts > q.max_retries:
                    self._ack_internal(q, msg)
                    self._dead_letter(
                        q, msg, error or f"max retries exceeded ({q.max_retries})", f.get("worker", "")
                    )
                else:
                    q.ready.appendleft(msg)
            self._kick(q)
            return None

        if method == "stats":
            name = f["queue"]
            q = self.queues.get(name)
            if q is None:
                raise KeyError(f"queue '{name}' not found")
            return {"stats": q.stats()}

        if method == "list":
            return {"queues": [q.stats() for q in self.queues.values()]}

        if method == "purge":
            q = self.queues.get(f["queue"])
            if q is None:
                return {"purged": 0}
            n = len(q.ready)
            if q.journal:
                for m in q.ready:
                    q.journal.append_ack(m.seq)
                self._dirty_journals.add(q.name)
            q.ready.clear()
            return {"purged": n}

        if method == "delete":
            q = self.queues.pop(f["queue"], None)
            if q is None:
                return {"deleted": False}
            for c in q.consumers:
                c.cancelled = True
            if q.journal:
                q.journal.delete()
            return {"deleted": True}

        if method == "peek":
            # Non-destructive read of up to `limit` ready messages (errors view).
            q = self.queues.get(f["queue"])
            limit = int(f.get("limit", 10))
            if q is None:
                return {"messages": []}
            return {"messages": [m.body for _, m in zip(range(limit), q.ready)]}

        if method == "heartbeat":
            wid = f["worker_id"]
            self.workers[wid] = {
                "worker_id": wid,
                "status": f.get("status", "active"),
                "last_seen": time.time(),
                "jobs_processed": f.get("jobs_processed", 0),
                "avg_duration_ms": f.get("avg_duration_ms"),
                "queue": f.get("queue", ""),
            }
            return None

        if method == "workers":
            return {"workers": list(self.workers.values())}

        raise ValueError(f"unknown method: {method}")


async def run_broker(
    host: str, port: int, data_dir: Optional[str], max_retries: int = 3
) -> None:
    server = BrokerServer(host, port, data_dir, max_retries)
    await server.serve()
    try:
        while True:
            await asyncio.sleep(3600)
    finally:
        await server.close()
