"""Logging setup (reference parity: llmq/utils/logging.py:8-75).

Workers log structured JSON to stdout (machine-parseable, `| jq .`);
CLI commands log human-readable lines to stderr.
"""

from __future__ import annotations

import json
import logging
import sys
from datetime import datetime, timezone
from typing import Optional


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        blob = {
            "timestamp": datetime.now(timezone.utc).isoformat(),
            "level": record.levelname,
            "logger": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info:
            blob["exception"] = self.formatException(record.exc_info)
        for key in ("worker_id", "job_id", "queue", "duration_ms"):
            value = getattr(record, key, None)
            if value is not None:
                blob[key] = value
        return json.dumps(blob, separators=(",", ":"), default=str)


def setup_logging(
    level: Optional[str] = None, json_output: bool = False, worker_id: Optional[str] = None
) -> None:
    from llmq_amd.core.config import get_config

    level_name = (level or get_config().log_level).upper()
    root = logging.getLogger()
    root.setLevel(getattr(logging, level_name, logging.INFO))
    for handler in list(root.handlers):
        root.removeHandler(handler)
    if json_output:
        handler = logging.StreamHandler(sys.stdout)
        handler.setFormatter(JsonFormatter())
    else:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(
            logging.Formatter("%(asctime)s %(levelname)-7s %(name)s: %(message)s")
        )
    if worker_id:
        old_factory = logging.getLogRecordFactory()

        def factory(*args, **kwargs):
            record = old_factory(*args, **kwargs)
            record.worker_id = worker_id
            return record

        logging.setLogRecordFactory(factory)
    root.addHandler(handler)
