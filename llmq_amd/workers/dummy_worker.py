"""CPU echo worker for tests and plumbing benchmarks.

Reference parity: llmq/workers/dummy_worker.py:9-51 — echoes the formatted
prompt after a configurable delay (default 1 s there; configurable here so
throughput tests of the broker itself can run at full speed).
"""

from __future__ import annotations

import asyncio
import uuid

from llmq_amd.core.models import Job
from llmq_amd.workers.base import BaseWorker


class DummyWorker(BaseWorker):
    def __init__(self, *args, delay_s: float = 1.0, **kwargs):
        super().__init__(*args, **kwargs)
        self.delay_s = delay_s

    def _generate_worker_id(self) -> str:
        return f"dummy-{uuid.uuid4().hex[:8]}"

    async def _initialize_processor(self) -> None:
        pass

    async def _process_job(self, job: Job) -> str:
        if self.delay_s > 0:
            await asyncio.sleep(self.delay_s)
        if job.messages is not None:
            text = " ".join(str(m.get("content", "")) for m in job.messages)
        else:
            text = job.get_formatted_prompt()
        return f"echo {text}"
