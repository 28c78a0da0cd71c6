"""Durable queue spool.

Append-only journal per queue: ``pub`` records add a message, ``ack``
records retire one. On startup the journal is replayed; messages without a
matching ack are restored ready-for-delivery (at-least-once, matching the
reference's durable-queue semantics at broker.py:70-78 / DeliveryMode
PERSISTENT 120-137, but owned by this process instead of RabbitMQ).

Compaction rewrites the file with only live messages once the retired
fraction is large.
"""

from __future__ import annotations

import json
import os
from pathlib import Path
from typing import Dict, Iterable, List, Tuple

_SAFE = set("abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789._-")


def safe_filename(queue_name: str) -> str:
    return "".join(c if c in _SAFE else f"%{ord(c):02x}" for c in queue_name)


class Journal:
    def __init__(self, directory: Path, queue_name: str, fsync: bool = False):
        self.path = directory / (safe_filename(queue_name) + ".jsonl")
        self.queue_name = queue_name
        self.fsync = fsync
        self._fh = None
        self._live = 0
        self._retired = 0

    # -- recovery --------------------------------------------------------

    def load(self) -> List[Tuple[int, str, str, int]]:
        """Replay → [(seq, msg_id, body, attempts)] still pending, seq-ordered."""
        pending: Dict[int, Tuple[int, str, str, int]] = {}
        if self.path.is_file():
            with open(self.path, "r", encoding="utf-8") as fh:
                for line in fh:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        rec = json.loads(line)
                    except json.JSONDecodeError:
                        continue  # torn tail write — drop
                    op = rec.get("op")
                    if op == "pub":
                        pending[rec["s"]] = (
                            rec["s"],
                            rec.get("id", ""),
                            rec["b"],
                            rec.get("a", 0),
                        )
                    elif op == "ack":
                        pending.pop(rec.get("s"), None)
        items = [pending[k] for k in sorted(pending)]
        self._open()
        self._live = len(items)
        self._retired = 0
        return items

    def _open(self) -> None:
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._fh = open(self.path, "a", encoding="utf-8")

    # -- mutation --------------------------------------------------------

    def append_publish(self, seq: int, msg_id: str, body: str, attempts: int = 0) -> None:
        if self._fh is None:
            self._open()
        self._fh.write(
            json.dumps(
                {"op": "pub", "s": seq, "id": msg_id, "b": body, "a": attempts},
                separators=(",", ":"),
            )
            + "\n"
        )
        self._live += 1

    def append_ack(self, seq: int) -> None:
        if self._fh is None:
            self._open()
        self._fh.write(json.dumps({"op": "ack", "s": seq}, separators=(",", ":")) + "\n")
        self._live -= 1
        self._retired += 1

    def flush(self) -> None:
        if self._fh is not None:
            self._fh.flush()
            if self.fsync:
                os.fsync(self._fh.fileno())

    def maybe_compact(self, live_messages: Iterable[Tuple[int, str, str, int]]) -> None:
        """Rewrite with only live messages when the file is mostly acks."""
        if self._retired < 10000 or self._retired < 2 * max(self._live, 1):
            return
        self.compact(live_messages)

    def compact(self, live_messages: Iterable[Tuple[int, str, str, int]]) -> None:
        tmp = self.path.with_suffix(".tmp")
        count = 0
        with open(tmp, "w", encoding="utf-8") as fh:
            for seq, msg_id, body, attempts in live_messages:
                fh.write(
                    json.dumps(
                        {"op": "pub", "s": seq, "id": msg_id, "b": body, "a": attempts},
                        separators=(",", ":"),
                    )
                    + "\n"
                )
                count += 1
            fh.flush()
            os.fsync(fh.fileno())
        if self._fh is not None:
            self._fh.close()
        os.replace(tmp, self.path)
        self._open()
        self._live = count
        self._retired = 0

    def delete(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None
        try:
            self.path.unlink()
        except FileNotFoundError:
            pass

    def close(self) -> None:
        if self._fh is not None:
            self.flush()
            self._fh.close()
            self._fh = None
