"""Semantic dedup / outlier / representative filtering worker.

Reference parity: llmq/workers/semhash_worker.py:10-191 — the reference
wraps the external ``semhash`` library. That library is not in this image,
so the worker carries an in-tree implementation: character n-gram hashing
vectors (a la HashingVectorizer) + cosine similarity for duplicate
detection, mean-distance for outliers/representatives. Batch-accumulating
like the reference (default batch_size 1000, semhash_worker.py:21).

Jobs pass through (result = the extracted text) when kept; filtered jobs
produce a result with ``filtered: true`` so downstream consumers can drop
them — same observable behavior as the reference's filter modes.
"""

from __future__ import annotations

import asyncio
import hashlib
import logging
import uuid
from typing import Dict, List, Optional, Tuple

import numpy as np

from llmq_amd.core.models import Job
from llmq_amd.workers.base import BaseWorker

logger = logging.getLogger(__name__)

_DIM = 1024


def _embed(text: str, n: int = 3) -> np.ndarray:
    """Hashed char-n-gram embedding, L2-normalised. Deterministic, no deps."""
    vec = np.zeros(_DIM, dtype=np.float32)
    t = f"  {text.lower()}  "
    for i in range(len(t) - n + 1):
        gram = t[i : i + n]
        h = int.from_bytes(hashlib.blake2b(gram.encode(), digest_size=8).digest(), "little")
        idx = h % _DIM
        sign = 1.0 if (h >> 63) & 1 else -1.0
        vec[idx] += sign
    norm = float(np.linalg.norm(vec))
    if norm > 0:
        vec /= norm
    return vec


class SemHashWorker(BaseWorker):
    """Modes: 'dedup' (default), 'outliers', 'representatives'."""

    def __init__(
        self,
        *args,
        mode: str = "dedup",
        batch_size: int = 1000,
        threshold: float = 0.9,
        text_field: Optional[str] = None,
        flush_interval_s: float = 5.0,
        **kwargs,
    ):
        super().__init__(*args, **kwargs)
        if mode not in ("dedup", "outliers", "representatives"):
            raise ValueError(f"unknown semhash mode: {mode}")
        self.mode = mode
        self.batch_size = batch_size
        self.threshold = threshold
        self.text_field = text_field
        self.flush_interval_s = flush_interval_s
        self._batch: List[Tuple[Job, str, asyncio.Future]] = []
        self._batch_lock: Optional[asyncio.Lock] = None
        self._flusher: Optional[asyncio.Task] = None
        self._kept = 0
        self._filtered = 0

    def _generate_worker_id(self) -> str:
        return f"semhash-{uuid.uuid4().hex[:8]}"

    async def _initialize_processor(self) -> None:
        self._batch_lock = asyncio.Lock()
        self._flusher = asyncio.create_task(self._flush_loop())

    async def _cleanup_processor(self) -> None:
        if self._flusher:
            self._flusher.cancel()
            try:
                await self._flusher
            except asyncio.CancelledError:
                pass
        await self._flush()

    def _extract_text(self, job: Job) -> str:
        if self.text_field:
            extra = job.extra_fields()
            if self.text_field in extra:
                return str(extra[self.text_field])
        if job.messages is not None:
            return " ".join(str(m.get("content", "")) for m in job.messages)
        return job.get_formatted_prompt()

    async def _process_job(self, job: Job) -> str:
        text = self._extract_text(job)
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        assert self._batch_lock is not None
        async with self._batch_lock:
            self._batch.append((job, text, fut))
            ready = len(self._batch) >= self.batch_size
        if ready:
            await self._flush()
        return await fut

    async def _flush_loop(self) -> None:
        while True:
            await asyncio.sleep(self.flush_interval_s)
            await self._flush()

    async def _flush(self) -> None:
        assert self._batch_lock is not None
        async with self._batch_lock:
            batch, self._batch = self._batch, []
        if not batch:
            return
        texts = [t for _, t, _ in batch]
        keep = self._filter_batch(texts)
        kept = int(sum(keep))
        self._kept += kept
        self._filtered += len(batch) - kept
        logger.info(
            "semhash %s batch: %d in, %d kept, %d filtered (totals %d/%d)",
            self.mode, len(batch), kept, len(batch) - kept, self._kept, self._filtered,
        )
        for (job, text, fut), keep_it in zip(batch, keep):
            if not fut.done():
                fut.set_result(text if keep_it else "")

    def _filter_batch(self, texts: List[str]) -> List[bool]:
        if not texts:
            return []
        emb = np.stack([_embed(t) for t in texts])  # [N, D], rows unit-norm
        if self.mode == "dedup":
            keep = [True] * len(texts)
            kept_rows: List[int] = []
            # Exact-duplicate fast path first.
            seen: Dict[str, int] = {}
            for i, t in enumerate(texts):
                if t in seen:
                    keep[i] = False
                    continue
                seen[t] = i
                if kept_rows:
                    sims = emb[kept_rows] @ emb[i]
                    if float(np.max(sims)) >= self.threshold:
                        keep[i] = False
                        continue
                kept_rows.append(i)
            return keep
        centroid = emb.mean(axis=0)
        cnorm = float(np.linalg.norm(centroid))
        if cnorm > 0:
            centroid = centroid / cnorm
        sims = emb @ centroid  # similarity to centroid
        if self.mode == "outliers":
            # Keep the least-central fraction (below the quantile).
            cutoff = float(np.quantile(sims, 0.1)) if len(texts) > 10 else float(np.min(sims))
            return [bool(s <= cutoff + 1e-9) for s in sims]
        # representatives: keep the most-central fraction
        cutoff = float(np.quantile(sims, 0.9)) if len(texts) > 10 else float(np.max(sims))
        return [bool(s >= cutoff - 1e-9) for s in sims]

    def _build_result(self, job: Job, output: str, duration_ms: float):
        result = super()._build_result(job, output, duration_ms)
        if output == "":
            # mark filtered-out rows so receivers can drop them
            data = result.model_dump()
            data["filtered"] = True
            result = type(result)(**data)
        return result
