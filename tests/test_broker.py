"""Broker + client integration tests (live in-process broker, real TCP).

Coverage modeled on reference tests/test_broker.py + test_integration.py,
but against the in-tree broker: publish/consume round-trip, prefetch
windows, ack/nack, DLQ with retry cap, durability across restart, stats,
purge, TTL.
"""

from __future__ import annotations

import asyncio
import json

import pytest

from llmq_amd.core.client import BrokerClient
from llmq_amd.core.models import Job
from tests.conftest import live_broker, run_async

pytestmark = pytest.mark.integration


def test_connect_and_ping():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            reply = await client.call({"m": "ping"})
            assert reply["pong"] is True
            await client.disconnect()

    run_async(main())


def test_connect_retry_fails_fast():
    async def main():
        from llmq_amd.core.config import Config

        client = BrokerClient(Config(broker_url="llmq://127.0.0.1:1"))
        with pytest.raises(ConnectionError):
            await client.connect(retries=2)

    run_async(main())


def test_publish_consume_roundtrip():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("q1")
            job = Job(id="j1", prompt="hello {x}", x="world")
            await client.publish_job("q1", job)

            got = asyncio.Event()
            bodies = []

            async def cb(delivery):
                bodies.append(delivery.body)
                await delivery.ack()
                got.set()

            await client.consume_jobs("q1", cb, prefetch=10)
            await asyncio.wait_for(got.wait(), 5)
            back = Job.model_validate_json(bodies[0])
            assert back.id == "j1"
            assert back.get_formatted_prompt() == "hello world"
            stats = await client.get_queue_stats("q1")
            assert stats.message_count == 0
            await client.disconnect()

    run_async(main())


def test_batch_publish_and_stats():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            jobs = [Job(id=f"j{i}", prompt=f"p{i}") for i in range(50)]
            await client.publish_jobs("qb", jobs)
            stats = await client.get_queue_stats("qb")
            assert stats.message_count == 50
            assert stats.message_count_ready == 50
            assert stats.message_bytes > 0
            assert stats.consumer_count == 0
            await client.disconnect()

    run_async(main())


def test_prefetch_window_caps_inflight():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            jobs = [Job(id=f"j{i}", prompt="p") for i in range(10)]
            await client.publish_jobs("qp", jobs)

            deliveries = []
            async def cb(delivery):
                deliveries.append(delivery)  # hold, don't ack

            await client.consume_jobs("qp", cb, prefetch=3)
            await asyncio.sleep(0.3)
            assert len(deliveries) == 3  # window full
            await deliveries[0].ack()
            await asyncio.sleep(0.3)
            assert len(deliveries) == 4  # one slot freed
            stats = await client.get_queue_stats("qp")
            assert stats.message_count_unacknowledged == 3
            await client.disconnect()

    run_async(main())


def test_nack_requeue_then_dlq_after_max_retries():
    async def main():
        async with live_broker(max_retries=2) as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.publish_job("qr", Job(id="poison", prompt="p"))

            attempts = []
            done = asyncio.Event()

            async def cb(delivery):
                attempts.append(delivery.attempts)
                await delivery.nack(requeue=True, error="boom", worker="w1")
                if len(attempts) >= 3:
                    done.set()

            await client.consume_jobs("qr", cb, prefetch=1)
            await asyncio.wait_for(done.wait(), 5)
            await asyncio.sleep(0.2)
            # attempts 1,2 requeued; attempt 3 > max_retries=2 → DLQ
            errors = await client.get_failed_messages("qr")
            assert len(errors) == 1
            assert errors[0].job_id == "poison"
            assert "boom" in errors[0].error_message
            stats = await client.get_queue_stats("qr")
            assert stats.message_count == 0
            await client.disconnect()

    run_async(main())


def test_nack_no_requeue_goes_straight_to_dlq():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.publish_job("qd", Job(id="bad", prompt="p"))
            done = asyncio.Event()

            async def cb(delivery):
                await delivery.nack(requeue=False, error="invalid", worker="w")
                done.set()

            await client.consume_jobs("qd", cb)
            await asyncio.wait_for(done.wait(), 5)
            await asyncio.sleep(0.2)
            errors = await client.get_failed_messages("qd")
            assert len(errors) == 1
            assert errors[0].error_message == "invalid"
            await client.disconnect()

    run_async(main())


def test_disconnect_requeues_unacked():
    async def main():
        async with live_broker() as (server, config):
            c1 = BrokerClient(config)
            await c1.connect()
            await c1.publish_job("qx", Job(id="j1", prompt="p"))

            held = asyncio.Event()

            async def hold(delivery):
                held.set()  # never ack

            await c1.consume_jobs("qx", hold, prefetch=1)
            await asyncio.wait_for(held.wait(), 5)
            await c1.disconnect()  # connection drop → requeue
            await asyncio.sleep(0.2)

            c2 = BrokerClient(config)
            await c2.connect()
            got = asyncio.Event()
            redelivered = []

            async def cb(delivery):
                redelivered.append(delivery.redelivered)
                await delivery.ack()
                got.set()

            await c2.consume_jobs("qx", cb)
            await asyncio.wait_for(got.wait(), 5)
            assert redelivered == [True]
            await c2.disconnect()

    run_async(main())


def test_round_robin_across_consumers():
    async def main():
        async with live_broker() as (server, config):
            c1 = BrokerClient(config)
            c2 = BrokerClient(config)
            await c1.connect()
            await c2.connect()
            n = 20
            await c1.publish_jobs("qrr", [Job(id=f"j{i}", prompt="p") for i in range(n)])

            seen1, seen2 = [], []
            done = asyncio.Event()

            def make_cb(seen):
                async def cb(delivery):
                    seen.append(json.loads(delivery.body)["id"])
                    await delivery.ack()
                    if len(seen1) + len(seen2) == n:
                        done.set()
                return cb

            await c1.consume_jobs("qrr", make_cb(seen1), prefetch=2)
            await c2.consume_jobs("qrr", make_cb(seen2), prefetch=2)
            await asyncio.wait_for(done.wait(), 10)
            assert len(seen1) + len(seen2) == n
            assert set(seen1) | set(seen2) == {f"j{i}" for i in range(n)}
            assert seen1 and seen2  # both got work
            await c1.disconnect()
            await c2.disconnect()

    run_async(main())


def test_durability_across_restart(tmp_path):
    async def main():
        from llmq_amd.broker.server import BrokerServer
        from llmq_amd.core.config import Config

        s1 = BrokerServer("127.0.0.1", 0, data_dir=str(tmp_path))
        await s1.serve()
        cfg1 = Config(broker_url=f"llmq://127.0.0.1:{s1.port}")
        c1 = BrokerClient(cfg1)
        await c1.connect()
        await c1.publish_jobs("qjournal", [Job(id=f"j{i}", prompt="p") for i in range(5)])
        await c1.disconnect()
        s1._flush_dirty()
        await s1.close()

        s2 = BrokerServer("127.0.0.1", 0, data_dir=str(tmp_path))
        await s2.serve()
        cfg2 = Config(broker_url=f"llmq://127.0.0.1:{s2.port}")
        c2 = BrokerClient(cfg2)
        await c2.connect()
        stats = await c2.get_queue_stats("qjournal")
        assert stats.message_count == 5

        got = []
        done = asyncio.Event()

        async def cb(delivery):
            got.append(json.loads(delivery.body)["id"])
            await delivery.ack()
            if len(got) == 5:
                done.set()

        await c2.consume_jobs("qjournal", cb)
        await asyncio.wait_for(done.wait(), 5)
        assert sorted(got) == [f"j{i}" for i in range(5)]
        await c2.disconnect()
        s2._flush_dirty()
        await s2.close()

        # acked messages must not reappear after another restart
        s3 = BrokerServer("127.0.0.1", 0, data_dir=str(tmp_path))
        await s3.serve()
        cfg3 = Config(broker_url=f"llmq://127.0.0.1:{s3.port}")
        c3 = BrokerClient(cfg3)
        await c3.connect()
        stats = await c3.get_queue_stats("qjournal")
        assert stats.message_count == 0
        await c3.disconnect()
        await s3.close()

    run_async(main())


def test_purge_and_results_queue_naming():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("qz")
            from llmq_amd.core.models import Result

            r = Result(id="a", prompt="p", result="o", worker_id="w", duration_ms=1.0)
            await client.publish_result("qz", r)
            stats = await client.get_queue_stats("qz.results")
            assert stats.message_count == 1
            purged = await client.clear_queue("qz.results")
            assert purged == 1
            await client.disconnect()

    run_async(main())


def test_ttl_expires_to_dlq():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.call({"m": "declare", "queue": "qttl", "ttl_ms": 50})
            await client.publish_job("qttl", Job(id="old", prompt="p"))
            await asyncio.sleep(0.15)

            async def cb(delivery):
                await delivery.ack()

            await client.consume_jobs("qttl", cb)
            await asyncio.sleep(0.3)
            errors = await client.get_failed_messages("qttl")
            assert len(errors) == 1
            assert "expired" in errors[0].error_message
            await client.disconnect()

    run_async(main())


def test_worker_heartbeat_registry():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.heartbeat("w1", "q", 10, 123.0)
            workers = await client.get_workers()
            assert workers[0]["worker_id"] == "w1"
            assert workers[0]["jobs_processed"] == 10
            await client.disconnect()

    run_async(main())


def test_crash_looping_consumer_dead_letters():
    """A job whose consumer DISCONNECTS (no nack) every delivery must hit
    the DLQ after max_retries — crash-loop poison jobs cannot cycle forever."""

    async def main():
        async with live_broker(max_retries=2) as (server, config):
            pub = BrokerClient(config)
            await pub.connect()
            await pub.setup_queue_infrastructure("crashq")
            await pub.publish_jobs("crashq", [Job(id="poison", prompt="x")])

            for _ in range(4):  # a few crash-loops
                victim = BrokerClient(config)
                await victim.connect()
                got = asyncio.Event()

                async def cb(delivery):
                    got.set()  # neither ack nor nack: simulate crash

                await victim.consume_jobs("crashq", cb, prefetch=1)
                try:
                    await asyncio.wait_for(got.wait(), 5)
                except asyncio.TimeoutError:
                    await victim.disconnect()
                    break  # no more deliveries: already dead-lettered
                await victim.disconnect()  # unacked → requeue, attempts += 1
                await asyncio.sleep(0.05)

            failed = await pub.get_failed_messages("crashq", limit=10)
            assert len(failed) == 1
            assert failed[0].job_id == "poison"
            stats = await pub.get_queue_stats("crashq")
            assert stats.message_count == 0  # gone from the main queue
            await pub.disconnect()

    run_async(main())


def test_bodies_with_newlines_and_unicode_roundtrip():
    """Frames are newline-delimited JSON: bodies containing raw newlines,
    unicode, and JSON metacharacters must survive the wire intact."""

    async def main():
        async with live_broker() as (server, config):
            c = BrokerClient(config)
            await c.connect()
            await c.setup_queue_infrastructure("wire")
            nasty = 'line1\nline2\t"quoted" \\backslash 雪 🚀 \r\n end'
            await c.publish_jobs("wire", [Job(id="n1", prompt=nasty)])
            got = []
            done = asyncio.Event()

            async def cb(delivery):
                got.append(Job.model_validate_json(delivery.body))
                await delivery.ack()
                done.set()

            await c.consume_jobs("wire", cb, prefetch=1)
            await asyncio.wait_for(done.wait(), 5)
            assert got[0].prompt == nasty
            await c.disconnect()

    run_async(main())


def test_journal_compaction_preserves_live_messages(tmp_path):
    """compact() must atomically rewrite the spool with only live messages;
    recovery after compaction restores exactly the unacked set."""
    from llmq_amd.broker.journal import Journal

    j = Journal(tmp_path, "cq")
    j.load()
    for i in range(50):
        j.append_publish(i + 1, f"m{i}", f"body-{i}")
    for i in range(40):  # ack the first 40
        j.append_ack(i + 1)
    j.flush()
    live = [(i + 1, f"m{i}", f"body-{i}", 0) for i in range(40, 50)]
    j.compact(iter(live))
    j.close()

    j2 = Journal(tmp_path, "cq")
    restored = j2.load()
    assert [(s, m, b) for s, m, b, _a in restored] == [
        (s, m, b) for s, m, b, _a in live
    ]
    # post-compaction appends still work and survive another reload
    j2.append_publish(99, "late", "late-body")
    j2.flush()
    j2.close()
    j3 = Journal(tmp_path, "cq")
    assert any(m == "late" for _s, m, _b, _a in j3.load())
    j3.close()
