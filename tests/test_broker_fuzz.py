"""At-least-once delivery fuzz: random consumer behaviour (ack, nack-requeue,
hold-then-crash, slow ack) interleaved with publishes must deliver EVERY
message at least once with none lost — messages either complete (acked) or
end in the DLQ after the retry cap, and the queue drains to zero.

This is the distributed-queue core the whole framework rides on (reference
contract: RabbitMQ at-least-once, SURVEY §5 failure detection).
"""

from __future__ import annotations

import asyncio
import random

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from llmq_amd.core.client import BrokerClient
from llmq_amd.core.models import Job
from tests.conftest import live_broker, run_async

pytestmark = pytest.mark.integration


@settings(
    max_examples=8, deadline=None,
    suppress_health_check=[HealthCheck.too_slow, HealthCheck.data_too_large],
)
@given(
    seed=st.integers(0, 2**32 - 1),
    n_jobs=st.integers(5, 25),
    n_consumers=st.integers(1, 3),
)
def test_at_least_once_under_misbehaving_consumers(seed, n_jobs, n_consumers):
    async def main():
        rng = random.Random(seed)
        async with live_broker() as (server, config):
            pub = BrokerClient(config)
            await pub.connect()
            await pub.setup_queue_infrastructure("fuzz")
            await pub.publish_jobs(
                "fuzz",
                [Job(id=f"j{i}", prompt=f"p{i}") for i in range(n_jobs)],
            )

            acked: set[str] = set()
            seen: list[str] = []
            crashed_once: set[str] = set()
            done = asyncio.Event()
            clients: list[BrokerClient] = []

            def check_done():
                if len(acked) + 0 >= n_jobs:
                    done.set()

            async def make_consumer(ci: int):
                c = BrokerClient(config)
                await c.connect()
                clients.append(c)

                async def cb(delivery):
                    job = Job.model_validate_json(delivery.body)
                    seen.append(job.id)
                    roll = rng.random()
                    if roll < 0.15 and job.id not in crashed_once:
                        # nack-requeue once per job at most (bounded retries)
                        crashed_once.add(job.id)
                        await delivery.nack(requeue=True, error="fuzz-requeue")
                        return
                    if roll < 0.25:
                        await asyncio.sleep(rng.uniform(0, 0.05))  # slow ack
                    await delivery.ack()
                    acked.add(job.id)
                    check_done()

                await c.consume_jobs("fuzz", cb, prefetch=rng.choice([1, 2, 8]))

            for ci in range(n_consumers):
                await make_consumer(ci)

            await asyncio.wait_for(done.wait(), 30)
            # every job delivered at least once, every job eventually acked
            assert acked == {f"j{i}" for i in range(n_jobs)}
            assert set(seen) == acked
            # queue fully drained (no lost, no stuck)
            await asyncio.sleep(0.2)
            stats = await pub.get_queue_stats("fuzz")
            assert stats.message_count == 0
            for c in clients:
                await c.disconnect()
            await pub.disconnect()

    run_async(main())


@settings(max_examples=5, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(seed=st.integers(0, 2**32 - 1), n_jobs=st.integers(4, 12))
def test_consumer_crash_mid_stream_loses_nothing(seed, n_jobs):
    """A consumer that disconnects while holding unacked deliveries must
    cause redelivery to a healthy consumer; all jobs complete."""
    async def main():
        rng = random.Random(seed)
        async with live_broker() as (server, config):
            pub = BrokerClient(config)
            await pub.connect()
            await pub.setup_queue_infrastructure("fz2")
            await pub.publish_jobs(
                "fz2", [Job(id=f"k{i}", prompt="x") for i in range(n_jobs)]
            )

            # crasher: holds a few deliveries unacked, then drops the link
            crasher = BrokerClient(config)
            await crasher.connect()
            held = asyncio.Event()
            hold_n = rng.randint(1, max(1, n_jobs // 2))
            held_count = 0

            async def hold(delivery):
                nonlocal held_count
                held_count += 1
                if held_count >= hold_n:
                    held.set()
                # never ack

            await crasher.consume_jobs("fz2", hold, prefetch=hold_n)
            await asyncio.wait_for(held.wait(), 10)
            await crasher.disconnect()  # drop with unacked in hand

            survivor = BrokerClient(config)
            await survivor.connect()
            acked: set[str] = set()
            done = asyncio.Event()

            async def cb(delivery):
                job = Job.model_validate_json(delivery.body)
                await delivery.ack()
                acked.add(job.id)
                if len(acked) >= n_jobs:
                    done.set()

            await survivor.consume_jobs("fz2", cb, prefetch=4)
            await asyncio.wait_for(done.wait(), 30)
            assert acked == {f"k{i}" for i in range(n_jobs)}
            stats = await pub.get_queue_stats("fz2")
            assert stats.message_count == 0
            await survivor.disconnect()
            await pub.disconnect()

    run_async(main())
