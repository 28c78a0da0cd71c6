"""Wire protocol for the in-tree broker.

Newline-delimited JSON frames over TCP. Requests carry a client-chosen
sequence number ``i``; the matching response echoes it. Unsolicited frames
(message deliveries) carry ``push`` instead.

Replaces the reference's AMQP 0-9-1 dependency (llmq/core/broker.py:5-11
uses aio-pika against an external RabbitMQ) with a self-contained protocol:

  client → server : {"i": 1, "m": "publish", "queue": "q", "body": "..."}
  server → client : {"i": 1, "ok": true}
  server → client : {"push": "deliver", "queue": "q", "tag": 7,
                     "body": "...", "redelivered": false, "attempts": 1}

Bodies are opaque UTF-8 strings (the Job/Result JSON from core.models).
"""

from __future__ import annotations

import asyncio
import json
from typing import Any, Dict

MAX_FRAME = 64 * 1024 * 1024  # 64 MiB — generous bound for batch publishes


def encode(frame: Dict[str, Any]) -> bytes:
    return json.dumps(frame, separators=(",", ":"), default=str).encode() + b"\n"


async def read_frame(reader: asyncio.StreamReader) -> Dict[str, Any]:
    """Read one frame; raises IncompleteReadError/LimitOverrunError on EOF."""
    line = await reader.readline()
    if not line:
        raise asyncio.IncompleteReadError(b"", None)
    if len(line) > MAX_FRAME:
        raise ValueError("frame too large")
    return json.loads(line)
