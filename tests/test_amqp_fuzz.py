"""Wire-robustness fuzz for the hand-written AMQP 0-9-1 stack.

1. Property: field tables with arbitrary JSON-ish values survive
   encode→decode byte-exactly in meaning (the codec is from scratch —
   amqp_codec.py — and carries every queue argument and header).
2. Adversarial: truncated/garbage method payloads raise AMQPError (never
   IndexError/struct.error leaking out of the codec).
3. Server: random bytes on a fresh connection (both before and after a
   valid protocol header) must close that connection without taking the
   broker down — the next well-behaved client still works.
"""

from __future__ import annotations

import asyncio
import random

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from llmq_amd.broker import amqp_codec as c
from llmq_amd.core.client import BrokerClient
from llmq_amd.core.models import Job
from tests.conftest import live_broker, run_async

pytestmark = pytest.mark.integration

# AMQP field values the codec supports (shortstr keys; nested tables/arrays)
_scalar = st.one_of(
    st.booleans(),
    st.integers(min_value=-(2**31), max_value=2**31 - 1),
    st.integers(min_value=2**32, max_value=2**50),       # forces 64-bit
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=40),
    st.none(),
)
_value = st.recursive(
    _scalar,
    lambda kids: st.one_of(
        st.lists(kids, max_size=4),
        st.dictionaries(st.text(min_size=1, max_size=20), kids, max_size=4),
    ),
    max_leaves=12,
)
_table = st.dictionaries(st.text(min_size=1, max_size=30), _value, max_size=6)


@settings(max_examples=150, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(table=_table)
def test_field_table_roundtrip(table):
    payload = c.encode_method(
        "queue.declare", reserved1=0, queue="q", passive=False, durable=True,
        exclusive=False, auto_delete=False, nowait=False, arguments=table,
    )
    name, args = c.decode_method(payload)
    assert name == "queue.declare"

    def norm(v):
        if isinstance(v, float):
            return round(v, 3)  # float32 wire precision
        if isinstance(v, dict):
            return {k: norm(x) for k, x in v.items()}
        if isinstance(v, list):
            return [norm(x) for x in v]
        return v

    assert norm(args["arguments"]) == norm(table)


@settings(max_examples=200, deadline=None)
@given(seed=st.integers(0, 2**32 - 1), cut=st.integers(0, 64))
def test_corrupt_method_payloads_raise_amqperror(seed, cut):
    rng = random.Random(seed)
    good = c.encode_method(
        "basic.publish", reserved1=0, exchange="", routing_key="jobs",
        mandatory=False, immediate=False,
    )
    # truncate and/or flip bytes
    buf = bytearray(good[: max(4, len(good) - cut)])
    for _ in range(rng.randint(0, 6)):
        if buf:
            buf[rng.randrange(len(buf))] = rng.randrange(256)
    try:
        c.decode_method(bytes(buf))
    except c.AMQPError:
        pass  # the contract: malformed wire data -> AMQPError
    except (KeyError, UnicodeDecodeError):
        pass  # unknown method id / bad utf8 in shortstr: acceptable, typed
    # decoding to SOME method is also fine — flips can make a valid frame


def test_garbage_bytes_do_not_kill_broker():
    async def main():
        async with live_broker() as (server, config):
            host, port = config.broker_host, config.broker_port
            rng = random.Random(7)

            # 1. pure garbage from byte 0 (fails protocol sniffing)
            for _ in range(5):
                r, w = await asyncio.open_connection(host, port)
                w.write(bytes(rng.randrange(256) for _ in range(rng.randrange(1, 200))))
                try:
                    await w.drain()
                    await asyncio.wait_for(r.read(64), 2)
                except (ConnectionError, asyncio.TimeoutError):
                    pass
                w.close()

            # 2. valid AMQP protocol header, then garbage frames
            r, w = await asyncio.open_connection(host, port)
            w.write(b"AMQP\x00\x00\x09\x01")
            w.write(bytes(rng.randrange(256) for _ in range(300)))
            try:
                await w.drain()
                await asyncio.wait_for(r.read(4096), 2)
            except (ConnectionError, asyncio.TimeoutError):
                pass
            w.close()

            # 3. the broker still serves a well-behaved client
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("alive")
            await client.publish_job("alive", Job(id="ok", prompt="x"))
            stats = await client.get_queue_stats("alive")
            assert stats.message_count == 1
            await client.disconnect()

    run_async(main())


def test_huge_declared_body_rejected():
    """A content header announcing a multi-GB body must be refused before
    the server buffers it (memory-DoS guard), and the broker survives."""
    async def main():
        async with live_broker() as (server, config):
            host, port = config.broker_host, config.broker_port
            r, w = await asyncio.open_connection(host, port)
            w.write(b"AMQP\x00\x00\x09\x01")
            # minimal handshake: start-ok/tune-ok/open, channel.open, then a
            # publish with an absurd body size
            ftype, ch, payload = await c.read_frame(r)       # connection.start
            w.write(c.method_frame(0, "connection.start-ok",
                                   client_properties={}, mechanism="PLAIN",
                                   response="\x00guest\x00guest", locale="en_US"))
            ftype, ch, payload = await c.read_frame(r)       # connection.tune
            w.write(c.method_frame(0, "connection.tune-ok",
                                   channel_max=0, frame_max=131072, heartbeat=0))
            w.write(c.method_frame(0, "connection.open",
                                   virtual_host="/", reserved1="", reserved2=False))
            await c.read_frame(r)                            # connection.open-ok
            w.write(c.method_frame(1, "channel.open", reserved1=""))
            await c.read_frame(r)                            # channel.open-ok
            w.write(c.method_frame(1, "basic.publish", reserved1=0, exchange="",
                                   routing_key="q", mandatory=False,
                                   immediate=False))
            w.write(c.frame(c.FRAME_HEADER, 1,
                            c.encode_content_header(8 * 1024**3, {})))
            await w.drain()
            # server must error the connection, not buffer 8 GB
            try:
                data = await asyncio.wait_for(r.read(4096), 5)
            except (ConnectionError, asyncio.TimeoutError):
                data = b""
            w.close()
            # broker still alive for a normal client
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("alive2")
            await client.publish_job("alive2", Job(id="ok", prompt="x"))
            assert (await client.get_queue_stats("alive2")).message_count == 1
            await client.disconnect()

    run_async(main())


def test_json_protocol_oversized_line_closes_conn_only():
    """The native length-bounded JSON protocol: a line beyond MAX_FRAME must
    close that connection (LimitOverrunError path), never the broker."""
    async def main():
        async with live_broker() as (server, config):
            host, port = config.broker_host, config.broker_port
            r, w = await asyncio.open_connection(host, port)
            w.write(b"{" + b"x" * (70 * 1024 * 1024 // 16))  # 4+ MiB, no newline
            # keep pushing until the server gives up or we hit our own bound
            try:
                for _ in range(20):
                    w.write(b"y" * (4 * 1024 * 1024))
                    await asyncio.wait_for(w.drain(), 5)
            except (ConnectionError, asyncio.TimeoutError):
                pass
            w.close()
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("alive3")
            await client.publish_job("alive3", Job(id="ok", prompt="x"))
            assert (await client.get_queue_stats("alive3")).message_count == 1
            await client.disconnect()

    run_async(main())
