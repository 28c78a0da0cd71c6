"""Multi-stage pipeline configuration.

Same YAML format and queue-name derivation as the reference
(llmq/core/pipeline.py:7-112): a pipeline is an ordered list of stages,
stage queues are ``pipeline.<name>.<stage>`` and final results land on
``pipeline.<name>.results``.

Beyond the reference: per-stage templates are honoured at EVERY stage (the
reference applies them only to the first stage at submit time —
broker.py:176-181 forwards result.result verbatim), and
per-stage sampling config (model, temperature, max_tokens, ...) is passed
to the worker instead of being ignored.
"""

from __future__ import annotations

import re
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml
from pydantic import BaseModel, Field, field_validator

_NAME_RE = re.compile(r"^[A-Za-z0-9_-]+$")


def _check_name(v: str, what: str) -> str:
    if not v or not isinstance(v, str) or not _NAME_RE.match(v):
        raise ValueError(
            f"{what} can only contain letters, numbers, hyphens, and underscores"
        )
    return v


class PipelineStage(BaseModel):
    name: str = Field(description="Stage name (unique within the pipeline)")
    worker: str = Field(description="Worker type: engine | dummy | semhash | filter")
    config: Dict[str, Any] = Field(default_factory=dict)

    @field_validator("name")
    @classmethod
    def _valid_name(cls, v: str) -> str:
        return _check_name(v, "Stage name")

    @property
    def model(self) -> Optional[str]:
        return self.config.get("model")

    @property
    def template(self) -> Optional[str]:
        # Stage prompt template, e.g. "Translate: {text}" or "{translation_result}"
        return self.config.get("template") or self.config.get("prompt")

    @property
    def messages(self):
        """Chat-template stage config (reference example-pipeline.yaml uses
        `messages:` lists with {var} placeholders instead of a string
        template). Returns the raw template list or None."""
        return self.config.get("messages")

    @property
    def result_field(self) -> str:
        """Extra-field name under which this stage's output is stored for the
        next stage's template (default: <stage>_result)."""
        return self.config.get("result_field", f"{self.name}_result")


class PipelineConfig(BaseModel):
    name: str
    stages: List[PipelineStage] = Field(min_length=1)
    config: Dict[str, Any] = Field(default_factory=dict)

    @field_validator("name")
    @classmethod
    def _valid_name(cls, v: str) -> str:
        return _check_name(v, "Pipeline name")

    @field_validator("stages")
    @classmethod
    def _unique_stages(cls, v: List[PipelineStage]) -> List[PipelineStage]:
        names = [s.name for s in v]
        if len(names) != len(set(names)):
            raise ValueError("All stage names must be unique within a pipeline")
        return v

    @classmethod
    def from_yaml_file(cls, path: Path | str) -> "PipelineConfig":
        path = Path(path)
        if not path.is_file():
            raise FileNotFoundError(f"Pipeline config not found: {path}")
        with open(path, "r", encoding="utf-8") as fh:
            data = yaml.safe_load(fh)
        if not isinstance(data, dict):
            raise ValueError(f"Invalid pipeline YAML: {path}")
        return cls(**data)

    # -- queue naming (reference parity: pipeline.py:82-112) -------------

    def get_stage_queue_name(self, stage_name: str) -> str:
        if stage_name not in [s.name for s in self.stages]:
            raise ValueError(f"Stage '{stage_name}' not found in pipeline '{self.name}'")
        return f"pipeline.{self.name}.{stage_name}"

    def get_pipeline_results_queue_name(self) -> str:
        return f"pipeline.{self.name}.results"

    def get_stage(self, stage_name: str) -> PipelineStage:
        for s in self.stages:
            if s.name == stage_name:
                return s
        raise ValueError(f"Stage '{stage_name}' not found in pipeline '{self.name}'")

    def get_next_stage(self, stage_name: str) -> Optional[PipelineStage]:
        names = [s.name for s in self.stages]
        idx = names.index(stage_name)
        return self.stages[idx + 1] if idx + 1 < len(self.stages) else None

    def is_last_stage(self, stage_name: str) -> bool:
        return self.stages[-1].name == stage_name

    def all_queue_names(self) -> List[str]:
        return [self.get_stage_queue_name(s.name) for s in self.stages] + [
            self.get_pipeline_results_queue_name()
        ]
