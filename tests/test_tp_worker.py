"""Tensor-parallel EngineWorker plumbing on CPU (gloo, world_size 2).

Exercises the full TP serving path the driver's 8-GPU tier uses — rank-0
worker + follower process, control broadcast over gloo, lockstep engine
replicas — with the "nccl"(RCCL) backend swapped for gloo (SURVEY §2.9:
multi-process CPU tests stand in for xGMI here).
"""

from __future__ import annotations

import os

import pytest

from llmq_amd.core.client import BrokerClient
from llmq_amd.core.models import Job
from llmq_amd.workers.engine_worker import EngineWorker
from tests.conftest import live_broker, run_async
from tests.test_workers import _collect_results, _start_worker, _stop_worker

pytestmark = [pytest.mark.integration, pytest.mark.slow]


@pytest.mark.timeout(180)
def test_engine_worker_tp2_gloo_roundtrip():
    os.environ["LLMQ_TP_MASTER_PORT"] = "29581"
    os.environ["MASTER_ADDR"] = "127.0.0.1"

    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("tpq")
            jobs = [
                Job(id=f"t{i}", prompt=f"hello {i}", max_tokens=4, temperature=0.0)
                for i in range(4)
            ]
            await client.publish_jobs("tpq", jobs)
            worker = EngineWorker(
                "tpq",
                model="tiny-llama",
                tensor_parallel_size=2,
                max_num_seqs=4,
                max_model_len=128,
                config=config,
                engine_overrides=dict(
                    device="cpu",
                    enforce_eager=True,
                    load_weights=False,
                    num_kv_blocks=128,
                    max_prefill_tokens=256,
                ),
            )
            task = await _start_worker(worker)
            results = await _collect_results(client, "tpq", 4, timeout=120.0)
            assert {r.id for r in results} == {f"t{i}" for i in range(4)}
            for r in results:
                assert r.output_tokens >= 1
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())
