"""Environment-driven configuration.

Same shape as the reference (llmq/core/config.py:9-69) with LLMQ_* names and
MI355X engine knobs added. ``.env`` files in the CWD are honoured without a
python-dotenv dependency (not in this image).
"""

from __future__ import annotations

import os
from pathlib import Path
from typing import Optional

from pydantic import BaseModel, Field

_ENV_LOADED = False


def load_dotenv(path: str = ".env") -> None:
    """Minimal .env loader: KEY=VALUE lines, '#' comments, no interpolation."""
    global _ENV_LOADED
    if _ENV_LOADED:
        return
    _ENV_LOADED = True
    p = Path(path)
    if not p.is_file():
        return
    try:
        for line in p.read_text().splitlines():
            line = line.strip()
            if not line or line.startswith("#") or "=" not in line:
                continue
            key, _, value = line.partition("=")
            key, value = key.strip(), value.strip().strip("'\"")
            os.environ.setdefault(key, value)
    except OSError:
        pass


# A user switching from the reference keeps a working .env: its names
# (reference config.py:10-44) are honoured as fallbacks for ours.
_REFERENCE_ALIASES = {
    "LLMQ_BROKER_URL": "RABBITMQ_URL",
    "LLMQ_QUEUE_PREFETCH": "VLLM_QUEUE_PREFETCH",
    "LLMQ_GPU_MEMORY_UTILIZATION": "VLLM_GPU_MEMORY_UTILIZATION",
    "LLMQ_MAX_NUM_SEQS": "VLLM_MAX_NUM_SEQS",
    "LLMQ_MAX_MODEL_LEN": "VLLM_MAX_MODEL_LEN",
    "LLMQ_MAX_TOKENS": "VLLM_MAX_TOKENS",
}


def _getenv(name: str) -> Optional[str]:
    value = os.getenv(name)
    if value is None and name in _REFERENCE_ALIASES:
        value = os.getenv(_REFERENCE_ALIASES[name])
    return value


def _env(name: str, default: str) -> str:
    value = _getenv(name)
    return default if value is None else value


def _env_int(name: str, default: int) -> int:
    value = _getenv(name)
    return default if value is None else int(value)


def _env_float(name: str, default: float) -> float:
    value = _getenv(name)
    return default if value is None else float(value)


def _env_opt_int(name: str) -> Optional[int]:
    value = _getenv(name)
    return int(value) if value else None


def _env_bool(name: str, default: bool) -> bool:
    value = _getenv(name)
    if value is None:
        return default
    return value.strip().lower() in ("1", "true", "yes", "on")


class Config(BaseModel):
    """Runtime configuration (reference parity: llmq/core/config.py)."""

    # -- Broker ----------------------------------------------------------
    broker_url: str = Field(
        default_factory=lambda: _env("LLMQ_BROKER_URL", "llmq://127.0.0.1:5672"),
        description="In-tree broker URL (llmq://host:port)",
    )
    broker_data_dir: str = Field(
        default_factory=lambda: _env("LLMQ_DATA_DIR", os.path.expanduser("~/.llmq/spool")),
        description="Durable queue spool directory for the broker process",
    )
    queue_prefetch: int = Field(
        default_factory=lambda: _env_int("LLMQ_QUEUE_PREFETCH", 100),
        description="Messages prefetched per worker (keeps engine admission full)",
    )
    max_retries: int = Field(
        default_factory=lambda: _env_int("LLMQ_MAX_RETRIES", 3),
        description="Delivery attempts before a job is dead-lettered",
    )
    job_ttl_minutes: int = Field(
        default_factory=lambda: _env_int("LLMQ_JOB_TTL_MINUTES", 30),
        description="Job TTL in minutes (0 disables)",
    )
    chunk_size: int = Field(
        default_factory=lambda: _env_int("LLMQ_CHUNK_SIZE", 10000),
        description="Jobs read/published per chunk at submit time",
    )
    log_level: str = Field(
        default_factory=lambda: _env("LLMQ_LOG_LEVEL", "INFO"),
    )

    # -- Engine ----------------------------------------------------------
    gpu_memory_utilization: float = Field(
        default_factory=lambda: _env_float("LLMQ_GPU_MEMORY_UTILIZATION", 0.9),
        description="Fraction of the MI355X's 288 GB HBM the engine may claim",
    )
    max_num_seqs: Optional[int] = Field(
        default_factory=lambda: _env_opt_int("LLMQ_MAX_NUM_SEQS"),
        description="Continuous-batching cap on concurrent sequences per step",
    )
    max_model_len: Optional[int] = Field(
        default_factory=lambda: _env_opt_int("LLMQ_MAX_MODEL_LEN"),
        description="Context window cap",
    )
    max_tokens: int = Field(
        default_factory=lambda: _env_int("LLMQ_MAX_TOKENS", 8192),
        description="Generation cap per request",
    )
    kv_block_size: int = Field(
        default_factory=lambda: _env_int("LLMQ_KV_BLOCK_SIZE", 16),
        description="Paged-KV block size (tokens per page)",
    )
    enable_hipgraph: bool = Field(
        default_factory=lambda: _env_bool("LLMQ_HIPGRAPH", True),
        description="Capture the decode step as a hipGraph",
    )
    temperature: float = Field(
        default_factory=lambda: _env_float("LLMQ_TEMPERATURE", 0.7),
        description="Default sampling temperature (reference hardcodes 0.7)",
    )

    @property
    def job_ttl_ms(self) -> int:
        return self.job_ttl_minutes * 60 * 1000

    @property
    def broker_host(self) -> str:
        return _parse_url(self.broker_url)[0]

    @property
    def broker_port(self) -> int:
        return _parse_url(self.broker_url)[1]


def _parse_url(url: str) -> tuple[str, int]:
    """Parse llmq://host:port or amqp://user:pass@host:port/vhost."""
    rest = url.split("://", 1)[-1]
    rest = rest.split("/", 1)[0]        # drop vhost/path
    rest = rest.rsplit("@", 1)[-1]      # drop credentials
    if ":" in rest:
        host, _, port = rest.rpartition(":")
        return host or "127.0.0.1", int(port)
    return rest or "127.0.0.1", 5672


def get_config() -> Config:
    load_dotenv()
    return Config()
