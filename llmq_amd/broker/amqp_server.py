"""AMQP 0-9-1 front-end for the in-tree broker.

The broker core (queues, journal durability, prefetch dispatch, DLQ, TTL —
``broker/server.py``) is protocol-agnostic; this module speaks AMQP 0-9-1
on the SAME listening port (the connection's first bytes disambiguate:
``AMQP\\x00\\x00\\x09\\x01`` vs a ``{``-prefixed JSON frame). That keeps the
reference's deployment story intact (north star: "AMQP (RabbitMQ) job/
result queues"; reference clients are aio-pika against :5672,
/root/reference/llmq/core/broker.py:27-49): AMQP tooling can point at this
broker, and the in-tree AMQPBrokerClient (core/amqp_client.py) can point at
either this broker or a real RabbitMQ.

Mapping onto the core:
- one AMQP *channel* = one broker ``Connection`` (delivery tags are
  per-channel in AMQP; per-Connection in the core — a 1:1 wrapper aligns
  them), so channel.close requeues that channel's unacked messages.
- default exchange only: publish routing_key == queue name (the only form
  the reference uses). exchange.declare / queue.bind are accepted no-ops.
- queue.declare arguments: ``x-message-ttl`` maps to the core's TTL
  dead-lettering. Dead-letter args are accepted but unnecessary — the core
  dead-letters rejects and retry-cap overflows to ``<q>.failed`` natively.
- publisher confirms (confirm.select) are supported: the enqueue is
  journaled before the basic.ack goes out.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Any, Dict, Optional, Tuple

from llmq_amd.broker import amqp_codec as c
from llmq_amd.broker.server import BrokerServer, Connection, Consumer

logger = logging.getLogger(__name__)

# confirm class (RabbitMQ extension)
c.METHODS[(85, 10)] = ("confirm.select", [("nowait", "b")])
c.METHODS[(85, 11)] = ("confirm.select-ok", [])
c.NAME_TO_ID["confirm.select"] = (85, 10)
c.NAME_TO_ID["confirm.select-ok"] = (85, 11)


class AMQPChannelConn(Connection):
    """Adapter: the broker core pushes deliveries through Connection.send;
    translate them into basic.deliver + content frames on our channel."""

    def __init__(self, session: "AMQPSession", channel: int):
        super().__init__(session.server, session.writer)
        self.session = session
        self.channel = channel
        self.qos_prefetch = 100
        self.confirm_mode = False
        self.publish_seq = 0
        self.ctag_by_consumer: Dict[int, str] = {}
        self.consumer_by_ctag: Dict[str, Consumer] = {}
        # in-progress publish: (routing_key, message_id) then header/body
        self.pending_pub: Optional[Tuple[str, Dict[str, Any]]] = None
        self.pending_size = 0
        self.pending_props: Dict[str, Any] = {}
        self.pending_body = b""

    async def send(self, frame: Dict[str, Any]) -> None:
        if self.closed or frame.get("push") != "deliver":
            return
        tag = frame["tag"]
        owner = self.tags.get(tag)
        ctag = ""
        if owner is not None:
            ctag = self.ctag_by_consumer.get(owner[1].consumer_id, "")
        body = frame["body"].encode("utf-8")
        props = {
            "delivery_mode": 2,
            "headers": {"x-attempts": int(frame.get("attempts", 1))},
        }
        await self.session.write(
            c.method_frame(
                self.channel, "basic.deliver", consumer_tag=ctag,
                delivery_tag=tag, redelivered=bool(frame.get("redelivered")),
                exchange="", routing_key=frame["queue"],
            )
            + c.content_frames(self.channel, body, props)
        )


class AMQPSession:
    def __init__(self, server: BrokerServer, reader: asyncio.StreamReader,
                 writer: asyncio.StreamWriter):
        self.server = server
        self.reader = reader
        self.writer = writer
        self.channels: Dict[int, AMQPChannelConn] = {}
        self._wlock = asyncio.Lock()
        self._hb_task: Optional[asyncio.Task] = None
        self.heartbeat = 0

    async def write(self, data: bytes) -> None:
        async with self._wlock:
            try:
                self.writer.write(data)
                await self.writer.drain()
            except (ConnectionError, RuntimeError):
                pass

    async def run(self) -> None:
        try:
            await self._handshake()
            while True:
                ftype, channel, payload = await c.read_frame(self.reader)
                if ftype == c.FRAME_HEARTBEAT:
                    continue
                if ftype == c.FRAME_METHOD:
                    name, args = c.decode_method(payload)
                    if await self._on_method(channel, name, args):
                        return  # connection closed cleanly
                elif ftype == c.FRAME_HEADER:
                    self._on_header(channel, payload)
                elif ftype == c.FRAME_BODY:
                    self._on_body(channel, payload)
        except (asyncio.IncompleteReadError, ConnectionError, c.AMQPError) as exc:
            logger.debug("amqp session ended: %r", exc)
        finally:
            if self._hb_task:
                self._hb_task.cancel()
            for conn in self.channels.values():
                conn.closed = True
                self.server._on_disconnect(conn)
            self.channels.clear()

    async def _handshake(self) -> None:
        # caller consumed the 8-byte protocol header already
        await self.write(c.method_frame(
            0, "connection.start", version_major=0, version_minor=9,
            server_properties={
                "product": "llmq-amd-broker", "version": "2.0",
                "capabilities": {
                    "basic.nack": True, "publisher_confirms": True,
                    "consumer_cancel_notify": True,
                },
            },
            mechanisms=b"PLAIN AMQPLAIN", locales=b"en_US",
        ))
        name, _args = await self._expect_method("connection.start-ok")
        await self.write(c.method_frame(
            0, "connection.tune", channel_max=2047,
            frame_max=c.DEFAULT_FRAME_MAX, heartbeat=60,
        ))
        name, args = await self._expect_method("connection.tune-ok")
        self.heartbeat = int(args.get("heartbeat", 0))
        name, _args = await self._expect_method("connection.open")
        await self.write(c.method_frame(0, "connection.open-ok", reserved1=""))
        if self.heartbeat > 0:
            self._hb_task = asyncio.create_task(self._hb_loop())

    async def _hb_loop(self) -> None:
        while True:
            await asyncio.sleep(max(1.0, self.heartbeat / 2))
            await self.write(c.heartbeat_frame())

    async def _expect_method(self, expected: str) -> Tuple[str, Dict[str, Any]]:
        while True:
            ftype, _ch, payload = await c.read_frame(self.reader)
            if ftype == c.FRAME_HEARTBEAT:
                continue
            if ftype != c.FRAME_METHOD:
                raise c.AMQPError(f"expected {expected}, got frame type {ftype}")
            name, args = c.decode_method(payload)
            if name != expected:
                raise c.AMQPError(f"expected {expected}, got {name}")
            return name, args

    def _chan(self, channel: int) -> AMQPChannelConn:
        conn = self.channels.get(channel)
        if conn is None:
            raise c.AMQPError(f"method on unopened channel {channel}")
        return conn

    async def _chan_error(self, channel: int, code: int, text: str) -> None:
        conn = self.channels.pop(channel, None)
        if conn is not None:
            conn.closed = True
            self.server._on_disconnect(conn)
        await self.write(c.method_frame(
            channel, "channel.close", reply_code=code, reply_text=text,
            class_id=0, method_id=0,
        ))

    # -- method dispatch --------------------------------------------------

    async def _on_method(self, channel: int, name: str, a: Dict[str, Any]) -> bool:
        s = self.server
        w = self.write
        if name == "connection.close":
            await w(c.method_frame(0, "connection.close-ok"))
            return True
        if name == "connection.close-ok":
            return True
        if name == "channel.open":
            self.channels[channel] = AMQPChannelConn(self, channel)
            await w(c.method_frame(channel, "channel.open-ok", reserved1=b""))
            return False
        if name == "channel.close":
            conn = self.channels.pop(channel, None)
            if conn is not None:
                conn.closed = True
                s._on_disconnect(conn)
            await w(c.method_frame(channel, "channel.close-ok"))
            return False
        if name == "channel.close-ok":
            return False

        conn = self._chan(channel)

        if name == "exchange.declare":
            await w(c.method_frame(channel, "exchange.declare-ok"))
            return False
        if name == "queue.bind":
            await w(c.method_frame(channel, "queue.bind-ok"))
            return False
        if name == "queue.declare":
            qname = a["queue"]
            if a.get("passive"):
                q = s.queues.get(qname)
                if q is None:
                    await self._chan_error(channel, 404, f"NOT_FOUND - no queue '{qname}'")
                    return False
            else:
                ttl = a.get("arguments", {}).get("x-message-ttl")
                q = s._declare(qname, durable=bool(a.get("durable", True)),
                               ttl_ms=int(ttl) if ttl is not None else None)
                s._kick(q)
            st = q.stats()
            await w(c.method_frame(
                channel, "queue.declare-ok", queue=qname,
                message_count=st["message_count_ready"],
                consumer_count=st["consumer_count"],
            ))
            return False
        if name == "queue.purge":
            q = s.queues.get(a["queue"])
            n = 0
            if q is not None:
                n = len(q.ready)
                if q.journal:
                    for m in q.ready:
                        q.journal.append_ack(m.seq)
                    s._dirty_journals.add(q.name)
                q.ready.clear()
            await w(c.method_frame(channel, "queue.purge-ok", message_count=n))
            return False
        if name == "queue.delete":
            q = s.queues.pop(a["queue"], None)
            n = 0
            if q is not None:
                n = len(q.ready)
                for cons in q.consumers:
                    cons.cancelled = True
                if q.journal:
                    q.journal.delete()
            await w(c.method_frame(channel, "queue.delete-ok", message_count=n))
            return False
        if name == "basic.qos":
            conn.qos_prefetch = max(1, int(a.get("prefetch_count") or 100))
            await w(c.method_frame(channel, "basic.qos-ok"))
            return False
        if name == "confirm.select":
            conn.confirm_mode = True
            if not a.get("nowait"):
                await w(c.method_frame(channel, "confirm.select-ok"))
            return False
        if name == "basic.consume":
            q = s._declare(a["queue"])
            cons = Consumer(conn, conn.next_consumer_id(), q.name, conn.qos_prefetch)
            conn.consumers[cons.consumer_id] = cons
            q.consumers.append(cons)
            ctag = a.get("consumer_tag") or f"ctag-{channel}-{cons.consumer_id}"
            conn.ctag_by_consumer[cons.consumer_id] = ctag
            conn.consumer_by_ctag[ctag] = cons
            if not a.get("nowait"):
                await w(c.method_frame(channel, "basic.consume-ok", consumer_tag=ctag))
            s._kick(q)
            return False
        if name == "basic.cancel":
            cons = conn.consumer_by_ctag.pop(a["consumer_tag"], None)
            if cons is not None:
                cons.cancelled = True
                conn.consumers.pop(cons.consumer_id, None)
                q = s.queues.get(cons.queue)
                if q and cons in q.consumers:
                    q.consumers.remove(cons)
            if not a.get("nowait"):
                await w(c.method_frame(channel, "basic.cancel-ok",
                                       consumer_tag=a["consumer_tag"]))
            return False
        if name == "basic.publish":
            conn.pending_pub = (a.get("routing_key", ""), a)
            return False
        if name == "basic.get":
            await self._basic_get(channel, conn, a)
            return False
        if name == "basic.ack":
            tags = sorted(t for t in conn.tags if t <= a["delivery_tag"]) \
                if a.get("multiple") else [a["delivery_tag"]]
            for t in tags:
                await s._dispatch(conn, "ack", {"tag": t})
            return False
        if name in ("basic.reject", "basic.nack"):
            tags = sorted(t for t in conn.tags if t <= a["delivery_tag"]) \
                if a.get("multiple") else [a["delivery_tag"]]
            for t in tags:
                await s._dispatch(conn, "nack", {
                    "tag": t, "requeue": bool(a.get("requeue")),
                    "error": "rejected (amqp)",
                })
            return False
        raise c.AMQPError(f"unhandled method {name}")

    async def _basic_get(self, channel: int, conn: AMQPChannelConn,
                         a: Dict[str, Any]) -> None:
        q = self.server.queues.get(a["queue"])
        if q is None or not q.ready:
            await self.write(c.method_frame(channel, "basic.get-empty", reserved1=""))
            return
        msg = q.ready.popleft()
        if a.get("no_ack"):
            self.server._ack_internal(q, msg)
            tag = conn.next_tag()
        else:
            tag = conn.next_tag()
            msg.attempts += 1
            q.unacked[(conn.id, tag)] = msg
            # hidden consumer so ack/nack bookkeeping works; not in
            # q.consumers, so _kick never dispatches to it
            cons = Consumer(conn, -1, q.name, 1, inflight=1)
            conn.tags[tag] = (q, cons)
        body = msg.body.encode("utf-8")
        await self.write(
            c.method_frame(
                channel, "basic.get-ok", delivery_tag=tag,
                redelivered=msg.attempts > 1, exchange="",
                routing_key=q.name, message_count=len(q.ready),
            )
            + c.content_frames(channel, body,
                               {"delivery_mode": 2, "message_id": msg.msg_id})
        )

    # -- content assembly -------------------------------------------------

    def _on_header(self, channel: int, payload: bytes) -> None:
        conn = self._chan(channel)
        if conn.pending_pub is None:
            raise c.AMQPError("content header without basic.publish")
        conn.pending_size, conn.pending_props = c.decode_content_header(payload)
        # Per-frame size is capped in read_frame, but the declared BODY size
        # spans many frames — without this bound a client announcing a huge
        # body makes the server buffer it all (memory DoS).
        if conn.pending_size > c.MAX_BODY_SIZE:
            raise c.AMQPError(
                f"message body too large: {conn.pending_size} "
                f"(max {c.MAX_BODY_SIZE})"
            )
        conn.pending_body = b""
        if conn.pending_size == 0:
            self._finish_publish(conn)

    def _on_body(self, channel: int, payload: bytes) -> None:
        conn = self._chan(channel)
        if conn.pending_pub is None:
            raise c.AMQPError("content body without basic.publish")
        conn.pending_body += payload
        if len(conn.pending_body) >= conn.pending_size:
            self._finish_publish(conn)

    def _finish_publish(self, conn: AMQPChannelConn) -> None:
        rkey, _a = conn.pending_pub  # default exchange: rkey == queue name
        props = conn.pending_props
        body = conn.pending_body.decode("utf-8", "replace")
        conn.pending_pub = None
        s = self.server
        q = s._declare(rkey)
        s._publish(q, body, props.get("message_id", ""))
        s._sync_if_required(q)  # fsync-before-confirm when enabled
        s._kick(q)
        if conn.confirm_mode:
            conn.publish_seq += 1
            t = asyncio.ensure_future(self.write(c.method_frame(
                conn.channel, "basic.ack", delivery_tag=conn.publish_seq,
                multiple=False,
            )))
            s._send_tasks.add(t)
            t.add_done_callback(s._send_tasks.discard)


async def handle_amqp_connection(server: BrokerServer,
                                 reader: asyncio.StreamReader,
                                 writer: asyncio.StreamWriter) -> None:
    """Entry point, called after the 8-byte protocol header was consumed."""
    session = AMQPSession(server, reader, writer)
    await session.run()
