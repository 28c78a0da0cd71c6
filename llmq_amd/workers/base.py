"""Worker lifecycle skeleton (reference parity: llmq/workers/base.py:15-275).

A worker consumes jobs from one queue, processes them, publishes results,
and acks ONLY after the result is published (at-least-once, base.py:212).
Error paths:
  - Job parse/validation error → ack-and-drop to the DLQ (poison job;
    reference ack-drops silently, base.py:228-235 — we dead-letter it so
    ``llmq errors`` can see it).
  - Processing error → nack(requeue=True); the broker dead-letters after
    max_retries (the reference requeues forever, base.py:245).

Pipeline workers additionally route: non-final stages publish the NEXT
stage's job with the next stage's template applied (fixing the reference's
stage-template gap, broker.py:176-181), final stages publish to
``pipeline.<name>.results``.
"""

from __future__ import annotations

import asyncio
import logging
import signal
import time
import uuid
from abc import ABC, abstractmethod
from typing import Any, Dict, Optional

from llmq_amd.core.client import BrokerClient, Delivery
from llmq_amd.core.config import Config, get_config
from llmq_amd.core.models import Job, Result
from llmq_amd.core.pipeline import PipelineConfig

logger = logging.getLogger(__name__)

HEARTBEAT_INTERVAL_S = 5.0


class BaseWorker(ABC):
    def __init__(
        self,
        queue_name: str,
        config: Optional[Config] = None,
        pipeline: Optional[PipelineConfig] = None,
        stage_name: Optional[str] = None,
        prefetch: Optional[int] = None,
    ):
        self.queue_name = queue_name
        self.config = config or get_config()
        self.pipeline = pipeline
        self.stage_name = stage_name
        self.prefetch = prefetch if prefetch is not None else self.config.queue_prefetch
        self.worker_id = self._generate_worker_id()
        self.client = BrokerClient(self.config)
        self.running = False
        self.jobs_processed = 0
        self.total_duration_ms = 0.0
        self._consumer_id: Optional[int] = None
        self._install_signal_handlers = True
        self._stop_event = asyncio.Event()

    # -- abstract surface (reference: base.py:58-75) ---------------------

    def _generate_worker_id(self) -> str:
        return f"{type(self).__name__.lower()}-{uuid.uuid4().hex[:8]}"

    @abstractmethod
    async def _initialize_processor(self) -> None: ...

    @abstractmethod
    async def _process_job(self, job: Job) -> str:
        """Process one job; returns the generated/produced text."""

    async def _cleanup_processor(self) -> None:  # noqa: B027
        pass

    # -- lifecycle -------------------------------------------------------

    async def initialize(self) -> None:
        await self._initialize_processor()
        await self.client.connect()
        if self.pipeline is not None and self.stage_name is not None:
            await self.client.setup_pipeline_infrastructure(self.pipeline)
            self.queue_name = self.pipeline.get_stage_queue_name(self.stage_name)
        else:
            await self.client.setup_queue_infrastructure(self.queue_name)
        self._consumer_id = await self.client.consume_jobs(
            self.queue_name, self._process_message, prefetch=self.prefetch
        )
        logger.info(
            "worker %s starting to consume from queue %s (prefetch=%d)",
            self.worker_id,
            self.queue_name,
            self.prefetch,
        )

    def _handle_signal(self, *_args: Any) -> None:
        logger.info("worker %s received shutdown signal", self.worker_id)
        self.running = False
        self._stop_event.set()

    async def run(self) -> None:
        if self._install_signal_handlers:
            loop = asyncio.get_event_loop()
            for sig in (signal.SIGINT, signal.SIGTERM):
                try:
                    loop.add_signal_handler(sig, self._handle_signal)
                except (NotImplementedError, RuntimeError):
                    pass
        await self.initialize()
        self.running = True
        last_beat = 0.0
        try:
            while self.running:
                now = time.time()
                if now - last_beat >= HEARTBEAT_INTERVAL_S:
                    last_beat = now
                    try:
                        await self.client.heartbeat(
                            self.worker_id,
                            self.queue_name,
                            self.jobs_processed,
                            self.avg_duration_ms,
                        )
                    except (ConnectionError, RuntimeError, asyncio.TimeoutError):
                        logger.warning("heartbeat failed; reconnecting")
                        try:
                            await self._reconnect()
                        except ConnectionError:
                            pass
                try:
                    await asyncio.wait_for(self._stop_event.wait(), timeout=1.0)
                except asyncio.TimeoutError:
                    pass
        finally:
            await self.cleanup()

    async def _reconnect(self) -> None:
        await self.client.disconnect()
        await self.client.connect()
        if self.pipeline is not None and self.stage_name is not None:
            await self.client.setup_pipeline_infrastructure(self.pipeline)
        else:
            await self.client.setup_queue_infrastructure(self.queue_name)
        self._consumer_id = await self.client.consume_jobs(
            self.queue_name, self._process_message, prefetch=self.prefetch
        )

    async def cleanup(self) -> None:
        self.running = False
        try:
            await self.client.heartbeat(
                self.worker_id, self.queue_name, self.jobs_processed,
                self.avg_duration_ms, status="stopped",
            )
        except Exception:
            pass
        await self._cleanup_processor()
        await self.client.disconnect()
        logger.info(
            "worker %s stopped after %d jobs", self.worker_id, self.jobs_processed
        )

    @property
    def avg_duration_ms(self) -> Optional[float]:
        if self.jobs_processed == 0:
            return None
        return self.total_duration_ms / self.jobs_processed

    # -- hot path --------------------------------------------------------

    async def _process_message(self, delivery: Delivery) -> None:
        start = time.perf_counter()
        try:
            job = Job.model_validate_json(delivery.body)
        except Exception as exc:  # malformed job: dead-letter, don't loop
            logger.warning("dropping malformed job: %s", exc)
            await delivery.nack(requeue=False, error=f"invalid job: {exc}", worker=self.worker_id)
            return
        try:
            output = await self._process_job(job)
        except ValueError as exc:
            # Deterministic bad input (reference ack-drops, base.py:228-235):
            # dead-letter so it is visible in `llmq errors`.
            logger.warning("job %s rejected: %s", job.id, exc)
            await delivery.nack(requeue=False, error=str(exc), worker=self.worker_id)
            return
        except Exception as exc:  # transient: requeue (broker caps retries)
            logger.exception("job %s failed", job.id)
            await delivery.nack(requeue=True, error=str(exc), worker=self.worker_id)
            return

        duration_ms = (time.perf_counter() - start) * 1000.0
        result = self._build_result(job, output, duration_ms)
        try:
            await self._publish_result(job, result)
            await delivery.ack()  # ack AFTER publish: at-least-once
        except (ConnectionError, RuntimeError) as exc:
            logger.error("publish failed for job %s: %s", job.id, exc)
            try:
                await delivery.nack(requeue=True, error=f"publish failed: {exc}", worker=self.worker_id)
            except Exception:
                pass
            return
        self.jobs_processed += 1
        self.total_duration_ms += duration_ms
        if self.jobs_processed % 100 == 0:
            logger.info(
                "worker %s: %d jobs, avg %.1f ms",
                self.worker_id,
                self.jobs_processed,
                self.avg_duration_ms or 0.0,
            )

    def _build_result(self, job: Job, output: str, duration_ms: float) -> Result:
        try:
            prompt = job.get_formatted_prompt() if job.prompt is not None else ""
        except Exception:
            prompt = job.prompt or ""
        extra = {
            k: v for k, v in job.extra_fields().items() if k not in Result.model_fields
        }
        return Result(
            id=job.id,
            prompt=prompt,
            result=output,
            worker_id=self.worker_id,
            duration_ms=duration_ms,
            **extra,
        )

    async def _publish_result(self, job: Job, result: Result) -> None:
        if self.pipeline is None or self.stage_name is None:
            await self.client.publish_result(self.queue_name, result)
            return
        stage = self.pipeline.get_stage(self.stage_name)
        if self.pipeline.is_last_stage(self.stage_name):
            await self.client.publish_to_queue(
                self.pipeline.get_pipeline_results_queue_name(),
                result.model_dump_json(),
                result.id,
            )
            return
        next_stage = self.pipeline.get_next_stage(self.stage_name)
        assert next_stage is not None
        data: Dict[str, Any] = dict(job.extra_fields())
        data[stage.result_field] = result.result
        template = next_stage.template
        if next_stage.messages is not None:
            # chat-template stage config (reference example-pipeline.yaml):
            # interpolate {vars} through the messages list
            from llmq_amd.utils.template import format_json_template

            next_job = Job(
                id=job.id,
                messages=format_json_template(next_stage.messages, data),
                **data,
            )
        elif template is not None:
            next_job = Job(id=job.id, prompt=template, **data)
        else:
            # No template on the next stage: forward the raw result text
            # (reference behavior, broker.py:176-181).
            next_job = Job(id=job.id, prompt=result.result, **data)
        await self.client.publish_job(
            self.pipeline.get_stage_queue_name(next_stage.name), next_job
        )
