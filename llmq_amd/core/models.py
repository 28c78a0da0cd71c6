"""Wire schema for jobs, results and stats.

JSON-compatible with the reference (llmq/core/models.py:6-90) so existing
job files / result consumers keep working, but written against pydantic v2
and extended with sampling parameters the reference hardcodes
(temperature 0.7 at vllm_worker.py:162) or ignores (per-stage temperature,
pipeline.py:132).
"""

from __future__ import annotations

from datetime import datetime, timezone
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict, Field, model_validator

_CONTROL_FIELDS = (
    "id",
    "prompt",
    "messages",
    "chat_mode",
    "stop",
    "temperature",
    "top_p",
    "top_k",
    "max_tokens",
    "seed",
)


def utcnow() -> datetime:
    return datetime.now(timezone.utc)


class Job(BaseModel):
    """One unit of inference work. Exactly one of prompt/messages."""

    model_config = ConfigDict(extra="allow")

    id: str = Field(..., description="Unique job identifier")
    prompt: Optional[str] = Field(None, description="Prompt template with {placeholders}")
    messages: Optional[List[Dict[str, Any]]] = Field(None, description="Chat messages")
    chat_mode: bool = Field(default=False, description="Apply the chat template")
    stop: Optional[List[str]] = Field(None, description="Stop sequences (None = EOS only)")
    # Per-job sampling overrides (engine falls back to worker/stage defaults).
    temperature: Optional[float] = Field(None)
    top_p: Optional[float] = Field(None)
    top_k: Optional[int] = Field(None)
    max_tokens: Optional[int] = Field(None)
    seed: Optional[int] = Field(None)

    @model_validator(mode="after")
    def _exactly_one_input(self) -> "Job":
        if self.prompt is not None and self.messages is not None:
            raise ValueError(
                "Cannot specify both 'prompt' and 'messages'. Use one or the other."
            )
        if self.prompt is None and self.messages is None:
            raise ValueError("Must specify either 'prompt' or 'messages'.")
        return self

    def extra_fields(self) -> Dict[str, Any]:
        """Template/data fields carried alongside the job (passthrough)."""
        return {
            k: v for k, v in self.model_dump().items() if k not in _CONTROL_FIELDS
        }

    def get_formatted_prompt(self) -> str:
        """Resolve {placeholders} in the prompt from the job's extra fields."""
        if self.prompt is None:
            raise ValueError("Cannot format prompt: prompt is None")
        data = self.extra_fields()
        try:
            return self.prompt.format(**data)
        except (KeyError, IndexError):
            # Prompts containing literal braces that are not placeholders for
            # provided fields are used verbatim.
            return self.prompt


class Result(BaseModel):
    model_config = ConfigDict(extra="allow")

    id: str = Field(..., description="Job ID this result corresponds to")
    prompt: str = Field(..., description="The formatted prompt that was processed")
    result: str = Field(..., description="Generated text")
    worker_id: str = Field(...)
    duration_ms: float = Field(...)
    timestamp: datetime = Field(default_factory=utcnow)
    # Engine-side observability (absent for dummy/CPU workers).
    queue_wait_ms: Optional[float] = Field(None, description="Broker-delivery → engine-admit")
    prefill_ms: Optional[float] = Field(None)
    decode_ms: Optional[float] = Field(None)
    prompt_tokens: Optional[int] = Field(None)
    output_tokens: Optional[int] = Field(None)
    finish_reason: Optional[str] = Field(None, description="stop | length | eos")


class QueueStats(BaseModel):
    queue_name: str
    message_count: Optional[int] = None
    message_count_ready: Optional[int] = None
    message_count_unacknowledged: Optional[int] = None
    consumer_count: Optional[int] = None
    message_bytes: Optional[int] = None
    message_bytes_ready: Optional[int] = None
    message_bytes_unacknowledged: Optional[int] = None
    processing_rate: Optional[float] = None
    stats_source: str = "broker"


class WorkerHealth(BaseModel):
    worker_id: str
    status: str
    last_seen: datetime
    jobs_processed: int
    avg_duration_ms: Optional[float] = None


class ErrorInfo(BaseModel):
    job_id: str
    error_message: str
    timestamp: datetime
    worker_id: Optional[str] = None
