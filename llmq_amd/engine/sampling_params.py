"""Per-request sampling parameters.

The reference hardcodes temperature 0.7 and exposes only max_tokens/stop
(vllm_worker.py:161-165). Here every job can override (core.models.Job
sampling fields)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class SamplingParams:
    temperature: float = 0.7
    top_p: float = 1.0
    top_k: int = 0  # 0 = disabled
    max_tokens: int = 8192
    stop: Optional[List[str]] = None
    seed: Optional[int] = None
    ignore_eos: bool = False

    def __post_init__(self) -> None:
        if self.temperature < 0:
            raise ValueError("temperature must be >= 0")
        if not 0 < self.top_p <= 1.0:
            raise ValueError("top_p must be in (0, 1]")
        if self.top_k < 0:
            raise ValueError("top_k must be >= 0")
        if self.max_tokens < 1:
            raise ValueError("max_tokens must be >= 1")

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0
