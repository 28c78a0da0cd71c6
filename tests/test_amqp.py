"""AMQP 0-9-1 backend tests.

Three layers:
1. Codec unit tests (frame/method/table/content round-trips) — these are
   the wire-format contract with real RabbitMQ.
2. Integration: the from-scratch AMQP client against the in-tree broker's
   AMQP front-end (same port as the JSON protocol, sniffed) — declare,
   publish+confirm, consume/ack, reject→DLQ, purge, passive-declare stats,
   failed-message peek, and a DummyWorker full stack over amqp://.
3. Against a real RabbitMQ when RABBITMQ_URL is set (skipped otherwise —
   mirrors reference tests/test_integration.py:19-21).
"""

from __future__ import annotations

import asyncio
import json
import os
import uuid

import pytest

from llmq_amd.broker import amqp_codec as c
from llmq_amd.core.amqp_client import AMQPBrokerClient
from llmq_amd.core.client import BrokerClient
from llmq_amd.core.config import Config
from llmq_amd.core.models import Job, Result
from tests.conftest import live_broker, run_async

pytestmark = pytest.mark.integration


# ----------------------------------------------------------------- codec --

class TestCodec:
    def test_method_roundtrip_bits_and_table(self):
        payload = c.encode_method(
            "queue.declare", reserved1=0, queue="jobs.q", passive=False,
            durable=True, exclusive=False, auto_delete=False, nowait=False,
            arguments={"x-message-ttl": 60000, "x-dead-letter-exchange": "",
                       "nested": {"a": True, "b": 1.5, "c": None},
                       "big": 2**40, "arr": [1, "two"]},
        )
        name, args = c.decode_method(payload)
        assert name == "queue.declare"
        assert args["queue"] == "jobs.q"
        assert args["durable"] is True and args["passive"] is False
        assert args["arguments"]["x-message-ttl"] == 60000
        assert args["arguments"]["nested"] == {"a": True, "b": 1.5, "c": None}
        assert args["arguments"]["big"] == 2**40
        assert args["arguments"]["arr"] == [1, "two"]

    def test_method_roundtrip_consecutive_bits(self):
        payload = c.encode_method(
            "basic.nack", delivery_tag=77, multiple=False, requeue=True)
        name, args = c.decode_method(payload)
        assert (name, args["delivery_tag"], args["multiple"], args["requeue"]) == (
            "basic.nack", 77, False, True)

    def test_content_header_roundtrip(self):
        props = {"delivery_mode": 2, "message_id": "m-1",
                 "headers": {"x-attempts": 3}, "content_type": "application/json"}
        size, out = c.decode_content_header(c.encode_content_header(1234, props))
        assert size == 1234
        assert out == props

    def test_frame_layout(self):
        f = c.method_frame(1, "basic.ack", delivery_tag=5, multiple=False)
        assert f[0] == c.FRAME_METHOD
        assert f[-1] == c.FRAME_END
        # channel
        assert int.from_bytes(f[1:3], "big") == 1

    def test_protocol_header_constant(self):
        assert c.PROTOCOL_HEADER == b"AMQP\x00\x00\x09\x01"


# ----------------------------------------------------- in-tree integration --

def _amqp_config(server) -> Config:
    return Config(broker_url=f"amqp://guest:guest@127.0.0.1:{server.port}/")


def test_factory_selects_backend():
    assert isinstance(BrokerClient(Config(broker_url="amqp://h:5672/")), AMQPBrokerClient)
    assert not isinstance(BrokerClient(Config(broker_url="llmq://h:5672")), AMQPBrokerClient)


def test_amqp_publish_consume_ack():
    async def main():
        async with live_broker() as (server, _cfg):
            client = BrokerClient(_amqp_config(server))
            assert isinstance(client, AMQPBrokerClient)
            await client.connect()
            await client.setup_queue_infrastructure("aq")
            await client.publish_job("aq", Job(id="j1", prompt="hi {x}", x="y"))
            await client.publish_jobs(
                "aq", [Job(id=f"j{i}", prompt="p") for i in range(2, 5)])

            got = []
            done = asyncio.Event()

            async def cb(d):
                got.append(json.loads(d.body)["id"])
                await d.ack()
                if len(got) == 4:
                    done.set()

            await client.consume_jobs("aq", cb, prefetch=2)
            await asyncio.wait_for(done.wait(), 10)
            assert sorted(got) == ["j1", "j2", "j3", "j4"]

            stats = await client.get_queue_stats("aq")
            assert stats.message_count == 0
            assert stats.consumer_count == 1
            assert stats.stats_source in ("amqp_fallback", "management_api")
            await client.disconnect()

    run_async(main())


def test_amqp_reject_dead_letters():
    async def main():
        async with live_broker() as (server, _cfg):
            client = BrokerClient(_amqp_config(server))
            await client.connect()
            await client.setup_queue_infrastructure("dq")
            await client.publish_job("dq", Job(id="poison", prompt="bad"))

            rejected = asyncio.Event()

            async def cb(d):
                await d.nack(requeue=False, error="boom")
                rejected.set()

            await client.consume_jobs("dq", cb)
            await asyncio.wait_for(rejected.wait(), 10)
            await asyncio.sleep(0.1)
            errors = await client.get_failed_messages("dq")
            assert len(errors) == 1
            assert errors[0].job_id == "poison"
            # peek is non-destructive
            errors2 = await client.get_failed_messages("dq")
            assert len(errors2) == 1
            await client.disconnect()

    run_async(main())


def test_amqp_results_roundtrip_and_purge():
    async def main():
        async with live_broker() as (server, _cfg):
            client = BrokerClient(_amqp_config(server))
            await client.connect()
            await client.setup_queue_infrastructure("rq")
            for i in range(3):
                await client.publish_result(
                    "rq", Result(id=f"r{i}", result=f"out{i}", prompt="p", duration_ms=1.0, worker_id="w"))
            stats = await client.get_queue_stats("rq.results")
            assert stats.message_count == 3
            purged = await client.clear_queue("rq.results")
            assert purged == 3
            stats = await client.get_queue_stats("rq.results")
            assert stats.message_count == 0
            await client.disconnect()

    run_async(main())


def test_amqp_missing_queue_stats_unavailable():
    async def main():
        async with live_broker() as (server, _cfg):
            client = BrokerClient(_amqp_config(server))
            await client.connect()
            stats = await client.get_queue_stats("never-declared")
            assert stats.stats_source == "unavailable"
            # channel recovers for the next operation
            await client.setup_queue_infrastructure("ok")
            stats = await client.get_queue_stats("ok")
            assert stats.stats_source in ("amqp_fallback", "management_api")
            await client.disconnect()

    run_async(main())


def test_amqp_redelivery_on_disconnect():
    async def main():
        async with live_broker() as (server, _cfg):
            c1 = BrokerClient(_amqp_config(server))
            await c1.connect()
            await c1.setup_queue_infrastructure("rd")
            await c1.publish_job("rd", Job(id="j1", prompt="p"))

            seen = asyncio.Event()

            async def cb_hold(d):
                seen.set()  # never acks

            await c1.consume_jobs("rd", cb_hold)
            await asyncio.wait_for(seen.wait(), 10)
            # drop the connection with the message unacked
            c1._writer.close()
            await asyncio.sleep(0.2)

            c2 = BrokerClient(_amqp_config(server))
            await c2.connect()
            got = asyncio.Event()
            attempts = []

            async def cb(d):
                attempts.append((d.redelivered, d.attempts))
                await d.ack()
                got.set()

            await c2.consume_jobs("rd", cb)
            await asyncio.wait_for(got.wait(), 10)
            assert attempts[0][0] is True  # redelivered flag set
            assert attempts[0][1] >= 2     # x-attempts header carried
            await c2.disconnect()

    run_async(main())


def test_dummy_worker_full_stack_over_amqp(tmp_path):
    """The reference's core loop (worker consumes, processes, publishes
    result, acks — base.py:137-245) over REAL AMQP framing end to end."""

    async def main():
        async with live_broker() as (server, _cfg):
            from llmq_amd.workers.dummy_worker import DummyWorker

            cfg = _amqp_config(server)
            submit = BrokerClient(cfg)
            await submit.connect()
            await submit.setup_queue_infrastructure("wq")
            for i in range(5):
                await submit.publish_job("wq", Job(id=f"job-{i}", prompt=f"text {i}"))

            worker = DummyWorker("wq", config=cfg, delay_s=0.0)
            wtask = asyncio.create_task(worker.run())

            results = []
            done = asyncio.Event()

            async def on_result(d):
                results.append(Result.model_validate_json(d.body))
                await d.ack()
                if len(results) == 5:
                    done.set()

            await submit.consume_results("wq", on_result)
            await asyncio.wait_for(done.wait(), 30)
            assert sorted(r.id for r in results) == [f"job-{i}" for i in range(5)]
            assert all(r.result.startswith("echo") for r in results)
            worker.running = False
            worker._stop_event.set()
            await asyncio.wait_for(wtask, 10)
            await submit.disconnect()

    run_async(main())


def test_amqp_large_body_multi_frame():
    """Bodies larger than frame_max must split into multiple body frames on
    publish and reassemble on both server and client (spec §2.3.7)."""

    async def main():
        async with live_broker() as (server, _cfg):
            client = BrokerClient(_amqp_config(server))
            await client.connect()
            await client.setup_queue_infrastructure("big")
            blob = "x" * 400_000 + "END"  # ~3 body frames at 128 KiB frame_max
            await client.publish_job("big", Job(id="big1", prompt=blob))

            got = asyncio.Event()
            bodies = []

            async def cb(d):
                bodies.append(d.body)
                await d.ack()
                got.set()

            await client.consume_jobs("big", cb)
            await asyncio.wait_for(got.wait(), 15)
            job = Job.model_validate_json(bodies[0])
            assert job.prompt == blob
            await client.disconnect()

    run_async(main())


# --------------------------------------------------------- real RabbitMQ --

@pytest.mark.rabbitmq
@pytest.mark.skipif(not os.environ.get("RABBITMQ_URL"),
                    reason="RABBITMQ_URL not set")
def test_against_real_rabbitmq():
    async def main():
        cfg = Config(broker_url=os.environ["RABBITMQ_URL"])
        client = BrokerClient(cfg)
        await client.connect()
        q = f"llmq-test-{uuid.uuid4().hex[:8]}"
        await client.setup_queue_infrastructure(q)
        await client.publish_job(q, Job(id="j1", prompt="hello"))
        got = asyncio.Event()
        bodies = []

        async def cb(d):
            bodies.append(d.body)
            await d.ack()
            got.set()

        await client.consume_jobs(q, cb)
        await asyncio.wait_for(got.wait(), 15)
        assert json.loads(bodies[0])["id"] == "j1"
        stats = await client.get_queue_stats(q)
        assert stats.stats_source == "amqp"
        for suffix in ("", ".results", ".failed"):
            await client.clear_queue(q + suffix)
        await client.disconnect()

    run_async(main())
