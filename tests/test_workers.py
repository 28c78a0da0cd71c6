"""Worker lifecycle + pipeline-routing tests against a live in-process broker.

Exceeds the reference's coverage (it tests only DummyWorker echo + one
integration round-trip; SURVEY §4 notes no pipeline-routing tests exist).
"""

from __future__ import annotations

import asyncio
import json

import pytest

from llmq_amd.core.client import BrokerClient
from llmq_amd.core.models import Job, Result
from llmq_amd.core.pipeline import PipelineConfig
from llmq_amd.workers.dummy_worker import DummyWorker
from llmq_amd.workers.semhash_worker import SemHashWorker
from tests.conftest import live_broker, run_async

pytestmark = pytest.mark.integration


async def _start_worker(worker):
    worker._install_signal_handlers = False
    task = asyncio.create_task(worker.run())
    for _ in range(100):
        if worker.running:
            break
        await asyncio.sleep(0.05)
    return task


async def _stop_worker(worker, task):
    worker.running = False
    worker._stop_event.set()
    try:
        await asyncio.wait_for(task, 5)
    except asyncio.TimeoutError:
        task.cancel()


async def _collect_results(client, queue, n, timeout=10.0):
    results = []
    done = asyncio.Event()

    async def cb(delivery):
        results.append(Result.model_validate_json(delivery.body))
        await delivery.ack()
        if len(results) >= n:
            done.set()

    await client.consume_results(queue, cb, prefetch=100)
    await asyncio.wait_for(done.wait(), timeout)
    return results


def test_dummy_worker_roundtrip():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("w1")
            await client.publish_jobs(
                "w1", [Job(id=f"j{i}", prompt="hi {name}", name=f"n{i}") for i in range(5)]
            )
            worker = DummyWorker("w1", config=config, delay_s=0)
            task = await _start_worker(worker)
            results = await _collect_results(client, "w1", 5)
            assert {r.id for r in results} == {f"j{i}" for i in range(5)}
            by_id = {r.id: r for r in results}
            assert by_id["j0"].result == "echo hi n0"
            assert by_id["j0"].prompt == "hi n0"
            # extra fields pass through to the result
            assert json.loads(by_id["j3"].model_dump_json())["name"] == "n3"
            assert worker.jobs_processed == 5
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())


def test_worker_acks_only_after_publish():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("w2")
            await client.publish_job("w2", Job(id="j1", prompt="p"))
            worker = DummyWorker("w2", config=config, delay_s=0)
            task = await _start_worker(worker)
            await _collect_results(client, "w2", 1)
            stats = await client.get_queue_stats("w2")
            assert stats.message_count == 0  # acked
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())


def test_malformed_job_dead_letters():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("w3")
            await client.publish_to_queue("w3", "this is not json", "raw1")
            # valid JSON but schema-invalid (neither prompt nor messages)
            await client.publish_to_queue("w3", json.dumps({"id": "x"}), "x")
            worker = DummyWorker("w3", config=config, delay_s=0)
            task = await _start_worker(worker)
            for _ in range(50):
                errors = await client.get_failed_messages("w3")
                if len(errors) >= 2:
                    break
                await asyncio.sleep(0.1)
            assert len(errors) == 2
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())


def test_worker_heartbeats_visible():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            worker = DummyWorker("w4", config=config, delay_s=0)
            task = await _start_worker(worker)
            await asyncio.sleep(0.1)
            workers = await client.get_workers()
            assert any(w["worker_id"] == worker.worker_id for w in workers)
            await _stop_worker(worker, task)
            workers = await client.get_workers()
            me = [w for w in workers if w["worker_id"] == worker.worker_id][0]
            assert me["status"] == "stopped"
            await client.disconnect()

    run_async(main())


PIPELINE_YAML = {
    "name": "mt",
    "stages": [
        {"name": "translate", "worker": "dummy",
         "config": {"template": "Translate: {text}"}},
        {"name": "format", "worker": "dummy",
         "config": {"template": "Format: {translate_result}"}},
    ],
}


def test_pipeline_two_stage_routing_with_templates():
    async def main():
        async with live_broker() as (server, config):
            pipeline = PipelineConfig(**PIPELINE_YAML)
            client = BrokerClient(config)
            await client.connect()
            await client.setup_pipeline_infrastructure(pipeline)

            w1 = DummyWorker("ignored", config=config, delay_s=0,
                             pipeline=pipeline, stage_name="translate")
            w2 = DummyWorker("ignored", config=config, delay_s=0,
                             pipeline=pipeline, stage_name="format")
            t1 = await _start_worker(w1)
            t2 = await _start_worker(w2)

            job = Job(id="p1", prompt="Translate: {text}", text="hallo")
            await client.publish_job(pipeline.get_stage_queue_name("translate"), job)

            results = []
            done = asyncio.Event()

            async def cb(delivery):
                results.append(Result.model_validate_json(delivery.body))
                await delivery.ack()
                done.set()

            await client.consume(pipeline.get_pipeline_results_queue_name(), cb)
            await asyncio.wait_for(done.wait(), 10)
            final = results[0]
            assert final.id == "p1"
            # stage1 echoes "echo Translate: hallo"; stage2 applies ITS template
            # to the stage1 result (the reference never does this — SURVEY §2)
            assert final.result == "echo Format: echo Translate: hallo"
            await _stop_worker(w1, t1)
            await _stop_worker(w2, t2)
            await client.disconnect()

    run_async(main())


def test_pipeline_queue_names():
    pipeline = PipelineConfig(**PIPELINE_YAML)
    assert pipeline.get_stage_queue_name("translate") == "pipeline.mt.translate"
    assert pipeline.get_pipeline_results_queue_name() == "pipeline.mt.results"
    assert pipeline.is_last_stage("format")
    assert pipeline.get_next_stage("translate").name == "format"
    with pytest.raises(ValueError):
        pipeline.get_stage_queue_name("nope")


def test_semhash_dedup_filters_duplicates():
    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("sh")
            texts = ["the quick brown fox jumps", "the quick brown fox jumps",
                     "a completely different sentence entirely", "the quick brown fox jumps!"]
            await client.publish_jobs(
                "sh",
                [Job(id=f"j{i}", prompt="{text}", text=t) for i, t in enumerate(texts)],
            )
            worker = SemHashWorker(
                "sh", config=config, batch_size=4, threshold=0.85, flush_interval_s=0.2
            )
            task = await _start_worker(worker)
            results = await _collect_results(client, "sh", 4)
            by_id = {r.id: r for r in results}
            assert by_id["j0"].result != ""
            assert by_id["j1"].result == ""  # exact dup filtered
            assert getattr(by_id["j1"], "filtered", False) is True
            assert by_id["j2"].result != ""  # distinct survives
            assert by_id["j3"].result == ""  # near-dup filtered
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())


def test_field_filter_worker():
    """`worker filter q field value`: matching jobs pass through, the rest
    are marked filtered (reference README.md:250, unimplemented there)."""
    from llmq_amd.workers.filter_worker import FieldFilterWorker

    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_queue_infrastructure("fq")
            await client.publish_jobs("fq", [
                Job(id="keep", prompt="hello", lang="nl"),
                Job(id="drop", prompt="bye", lang="en"),
                Job(id="nofield", prompt="zz"),
            ])
            worker = FieldFilterWorker("fq", field="lang", value="nl", config=config)
            task = await _start_worker(worker)
            results = await _collect_results(client, "fq", 3, timeout=30.0)
            by_id = {r.id: r for r in results}
            assert by_id["keep"].result == "hello"
            assert not getattr(by_id["keep"], "filtered", False)
            for rid in ("drop", "nofield"):
                assert getattr(by_id[rid], "filtered", False) is True, by_id[rid]
                assert by_id[rid].result == ""
            await _stop_worker(worker, task)
            await client.disconnect()

    run_async(main())


def test_pipeline_messages_stage_templates(tmp_path):
    """The reference's example-pipeline.yaml uses `messages:` chat templates
    in stage config (never applied there — SURVEY §2 quirks). Here: the
    first stage's messages interpolate per row at submit, and stage N>1
    messages interpolate the previous stage's result."""
    import json as _json

    yaml_text = """
name: msgs-pipe
stages:
  - name: first
    worker: dummy
    config:
      messages:
        - role: user
          content: "Translate {source_lang}: {source_text}"
  - name: second
    worker: dummy
    config:
      messages:
        - role: user
          content: "Clean up: {first_result}"
"""
    cfg_path = tmp_path / "p.yaml"
    cfg_path.write_text(yaml_text)
    from llmq_amd.core.pipeline import PipelineConfig
    from llmq_amd.workers.dummy_worker import DummyWorker

    pipeline = PipelineConfig.from_yaml_file(str(cfg_path))

    async def main():
        async with live_broker() as (server, config):
            client = BrokerClient(config)
            await client.connect()
            await client.setup_pipeline_infrastructure(pipeline)

            # submit through the PipelineSubmitter mapping path
            from llmq_amd.cli.submit import PipelineSubmitter

            src = tmp_path / "rows.jsonl"
            src.write_text(_json.dumps(
                {"id": "m1", "source_lang": "nl", "source_text": "hallo"}) + "\n")
            import llmq_amd.core.config as cfg_mod
            old_get = cfg_mod.get_config
            cfg_mod.get_config = lambda: config
            try:
                sub = PipelineSubmitter(pipeline, str(src))
            finally:
                cfg_mod.get_config = old_get
            sub.config = config
            sub.client = BrokerClient(config)
            await sub.client.connect()
            rows = list(sub._rows())
            job = sub._job_from_row(rows[0], 0) if hasattr(sub, "_job_from_row") else None
            if job is None:
                from llmq_amd.utils.template import create_job_from_data
                job = create_job_from_data(rows[0], "m1", sub.template, sub.column_mapping)
            assert job.messages == [
                {"role": "user", "content": "Translate nl: hallo"}]
            await sub.client.publish_job(
                pipeline.get_stage_queue_name("first"), job)
            await sub.client.disconnect()

            w1 = DummyWorker(pipeline.get_stage_queue_name("first"),
                             delay_s=0, config=config, pipeline=pipeline,
                             stage_name="first")
            w2 = DummyWorker(pipeline.get_stage_queue_name("second"),
                             delay_s=0, config=config, pipeline=pipeline,
                             stage_name="second")
            t1 = await _start_worker(w1)
            t2 = await _start_worker(w2)

            got = asyncio.Event()
            final = []

            async def cb(delivery):
                final.append(Result.model_validate_json(delivery.body))
                await delivery.ack()
                got.set()

            await client.consume(
                pipeline.get_pipeline_results_queue_name(), cb, prefetch=5)
            await asyncio.wait_for(got.wait(), 30)
            # stage 1 echoed the chat content; stage 2's messages saw it
            assert final[0].result == "echo Clean up: echo Translate nl: hallo"
            await _stop_worker(w1, t1)
            await _stop_worker(w2, t2)
            await client.disconnect()

    run_async(main())
