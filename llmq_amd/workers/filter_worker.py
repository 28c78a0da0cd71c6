"""Field-equality filter worker.

The reference README documents ``llmq worker filter <queue> <field> <value>``
(README.md:250) but never implements it (cli/worker.py supports only
vllm/dummy/semhash — SURVEY §2 quirks). Implemented here with the semantics
that fit the documented CLI: a job whose ``field`` equals ``value`` passes
through (result = its text / field value); a non-matching job produces a
result marked ``filtered: true`` so ``llmq receive --skip-filtered`` (and
downstream pipeline stages) can drop it. Values compare as strings so CLI
usage like ``worker filter q lang nl`` matches ``{"lang": "nl"}``.
"""

from __future__ import annotations

import uuid

from llmq_amd.core.models import Job
from llmq_amd.workers.base import BaseWorker


class FieldFilterWorker(BaseWorker):
    def __init__(self, *args, field: str, value: str, **kwargs):
        # before super().__init__: the base constructor calls
        # _generate_worker_id(), which reads self.field
        self.field = field
        self.value = value
        super().__init__(*args, **kwargs)

    def _generate_worker_id(self) -> str:
        return f"filter-{self.field}-{uuid.uuid4().hex[:8]}"

    async def _initialize_processor(self) -> None:
        pass

    async def _process_job(self, job: Job) -> str:
        got = job.model_dump().get(self.field)
        if got is None:
            got = (job.model_extra or {}).get(self.field)
        if got is not None and str(got) == self.value:
            if job.messages is not None:
                return " ".join(str(m.get("content", "")) for m in job.messages)
            return job.get_formatted_prompt()
        return ""  # filtered (marked in _build_result)

    def _build_result(self, job: Job, output: str, duration_ms: float):
        result = super()._build_result(job, output, duration_ms)
        if output == "":
            data = result.model_dump()
            data["filtered"] = True
            result = type(result)(**data)
        return result
