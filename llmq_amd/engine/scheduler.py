"""Continuous-batching scheduler.

The in-tree equivalent of the capability the reference buys from vLLM
(admit up to max_num_seqs sequences per step, prefill/decode
interleave, paged-KV admission control). Policy:

- One step = the DECODE batch (every running sequence) plus a PREFILL
  segment built within the max_prefill_tokens budget ("mixed" step, one
  forward with per-segment attention) — decode never stalls behind new
  prompts. The broker-side prefetch ≫ batch trick keeps the
  waiting queue fed.
- The prefill segment first CONTINUES chunked long prompts, then admits
  new sequences FCFS; a prompt larger than the budget prefills chunk by
  chunk against the paged cache (mid-chunk sequences live in
  ``prefilling`` and sample no tokens until their context completes).
- Preemption by recompute: if a decode step cannot allocate the next KV
  block, the youngest running sequence is evicted, its blocks freed, and it
  re-enters the waiting queue with its generated tokens as prompt.
"""

from __future__ import annotations

import itertools
import logging
from collections import deque
from dataclasses import dataclass, field
from enum import Enum
from typing import Deque, List, Optional

from llmq_amd.engine.kv_cache import BlockAllocator, blocks_needed
from llmq_amd.engine.sampling_params import SamplingParams

logger = logging.getLogger(__name__)


class SeqStatus(Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"


_SEQ_UID = itertools.count()


class Sequence:
    __slots__ = (
        "request_id", "token_ids", "prompt_len", "params", "status",
        "block_table", "arrival_time", "first_token_time", "finish_reason",
        "output_text", "num_preemptions", "prefilled", "prefill_start_time",
        "uid",
    )

    def __init__(self, request_id: str, prompt_token_ids: List[int], params: SamplingParams,
                 arrival_time: float = 0.0):
        # Monotonic per-process id: request_ids may be REUSED across sequence
        # lifetimes (generate_batch reuses "batch-{i}"), so anything caching
        # per-sequence state (e.g. the hipGraph block-table staging rows)
        # must key on uid, not request_id.
        self.uid = next(_SEQ_UID)
        self.request_id = request_id
        self.token_ids: List[int] = list(prompt_token_ids)
        self.prompt_len = len(prompt_token_ids)
        self.params = params
        self.status = SeqStatus.WAITING
        self.block_table: List[int] = []
        self.arrival_time = arrival_time
        self.first_token_time: Optional[float] = None
        self.finish_reason: Optional[str] = None
        self.output_text = None  # set only when a stop string truncates
        self.num_preemptions = 0
        self.prefilled = 0  # context tokens whose KV is in the cache
        self.prefill_start_time: Optional[float] = None

    @property
    def num_tokens(self) -> int:
        return len(self.token_ids)

    @property
    def output_len(self) -> int:
        return len(self.token_ids) - self.prompt_len

    def append_token(self, token_id: int) -> None:
        self.token_ids.append(token_id)


@dataclass
class ScheduledBatch:
    kind: str  # "prefill" | "decode" | "mixed"
    seqs: List[Sequence] = field(default_factory=list)
    # mixed: seqs[:n_decode] are decoding, seqs[n_decode:] are prefilling
    n_decode: int = 0
    # per prefill entry: (start, end) token range entering the cache this
    # step; end < num_tokens ⇒ a chunk of a long prompt (no sampling yet)
    chunks: List[tuple] = field(default_factory=list)

    @property
    def empty(self) -> bool:
        return not self.seqs


class Scheduler:
    def __init__(
        self,
        allocator: BlockAllocator,
        block_size: int,
        max_num_seqs: int,
        max_prefill_tokens: int,
        max_model_len: int,
    ):
        self.allocator = allocator
        self.block_size = block_size
        self.max_num_seqs = max_num_seqs
        self.max_prefill_tokens = max_prefill_tokens
        self.max_model_len = max_model_len
        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []
        # long prompts being prefilled chunk-by-chunk (not yet decodable)
        self.prefilling: List[Sequence] = []

    # -- public ----------------------------------------------------------

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running) + len(self.prefilling)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running or self.prefilling)

    def add(self, seq: Sequence) -> None:
        if seq.num_tokens > self.max_model_len:
            # Trim oversized prompts from the left (keep the recent window).
            seq.token_ids = seq.token_ids[-self.max_model_len + 1 :]
            seq.prompt_len = len(seq.token_ids)
        # Reject a prompt the pool could NEVER seat even when empty —
        # otherwise it blocks the head of the waiting queue forever and the
        # engine livelocks on empty steps (has_work() stays true while no
        # batch can ever be built).
        need = blocks_needed(seq.num_tokens, self.block_size)
        if need > self.allocator.num_blocks:
            raise ValueError(
                f"request {seq.request_id}: prompt needs {need} KV blocks but "
                f"the pool has only {self.allocator.num_blocks} — it can never "
                "be admitted (raise gpu_memory_utilization/num_kv_blocks or "
                "lower max_model_len)"
            )
        self.waiting.append(seq)

    def abort(self, request_id: str) -> bool:
        for pool in (self.running, self.prefilling):
            for i, seq in enumerate(pool):
                if seq.request_id == request_id:
                    self._release(seq)
                    del pool[i]
                    return True
        for i, seq in enumerate(self.waiting):
            if seq.request_id == request_id:
                del self.waiting[i]
                return True
        return False

    def finish(self, seq: Sequence, reason: str) -> None:
        seq.status = SeqStatus.FINISHED
        seq.finish_reason = reason
        self._release(seq)
        try:
            self.running.remove(seq)
        except ValueError:
            pass

    def schedule(self) -> ScheduledBatch:
        # One step = the decode batch (all running seqs) + a prefill segment
        # built from (a) continuing chunks of long prompts, (b) new
        # admissions, within max_prefill_tokens — decode never stalls.
        decode = self._schedule_decode() if self.running else ScheduledBatch("decode")
        prefills, chunks = self._build_prefill_segment(len(decode.seqs))
        if not prefills:
            return decode
        if decode.empty:
            return ScheduledBatch("prefill", prefills, chunks=chunks)
        return ScheduledBatch(
            "mixed", decode.seqs + prefills, n_decode=len(decode.seqs),
            chunks=chunks,
        )

    def _build_prefill_segment(self, seats_used: int):
        prefills: List[Sequence] = []
        chunks: List[tuple] = []
        budget = self.max_prefill_tokens
        bs = self.block_size
        # (a) continue chunked prompts first (FCFS among them)
        for seq in list(self.prefilling):
            if budget <= 0:
                break
            start = seq.prefilled
            end = min(seq.num_tokens, start + budget)
            need = blocks_needed(end, bs) - len(seq.block_table)
            if need > 0:
                blocks = self.allocator.allocate(need)
                if blocks is None:
                    break
                seq.block_table.extend(blocks)
            prefills.append(seq)
            chunks.append((start, end))
            budget -= end - start
        # (b) admit new sequences. Seats are held not just by this step's
        # prefill segment but also by chunked sequences STRANDED this step
        # (budget/allocation ran out above) — they still occupy KV blocks
        # and will rejoin a later segment, so count them or running+prefilling
        # can exceed max_num_seqs (oversized decode batches then silently
        # fall off the hipGraph path).
        stranded = sum(1 for s in self.prefilling if s not in prefills)
        while (
            self.waiting and budget > 0
            and seats_used + len(prefills) + stranded < self.max_num_seqs
        ):
            seq = self.waiting[0]
            n = seq.num_tokens
            end = min(n, budget)
            if end < n and prefills:
                break  # start a long prompt's first chunk only at segment head
            blocks = self.allocator.allocate(blocks_needed(end, bs))
            if blocks is None:
                break
            seq.block_table = blocks
            seq.status = SeqStatus.RUNNING
            self.waiting.popleft()
            prefills.append(seq)
            chunks.append((0, end))
            budget -= end
            if end < n:
                break  # chunked head consumed the budget
        # move bookkeeping: where does each prefill seq live after this step?
        for seq, (start, end) in zip(prefills, chunks):
            if seq in self.prefilling:
                if end == seq.num_tokens:
                    self.prefilling.remove(seq)
                    self.running.append(seq)
            elif end < seq.num_tokens:
                self.prefilling.append(seq)
            else:
                self.running.append(seq)
        return prefills, chunks

    # -- internals -------------------------------------------------------

    def _release(self, seq: Sequence) -> None:
        if seq.block_table:
            self.allocator.free(seq.block_table)
            seq.block_table = []


    def _schedule_decode(self) -> ScheduledBatch:
        batch = ScheduledBatch("decode")
        if not self.running:
            return batch
        # Ensure every running seq has a KV slot for its next position;
        # preempt from the back (youngest) on allocation failure.
        # Cheap gate first: a decode step grows a seq by one token, so a new
        # block is needed only when the tokens exceed the table's capacity —
        # ~1/block_size of seqs per step take the slow path.
        bs = self.block_size
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            if len(seq.token_ids) <= len(seq.block_table) * bs:
                i += 1
                continue
            need = blocks_needed(seq.num_tokens, self.block_size) - len(seq.block_table)
            if need > 0:
                blocks = self.allocator.allocate(need)
                if blocks is None:
                    victim = self.running[-1]
                    if victim is seq and len(self.running) == 1:
                        logger.error(
                            "seq %s cannot fit in KV cache even alone; finishing as length",
                            seq.request_id,
                        )
                        self.finish(seq, "length")
                        continue
                    self._preempt(victim)
                    continue  # retry same index
                seq.block_table.extend(blocks)
            i += 1
        batch.seqs = list(self.running)
        return batch

    def _preempt(self, seq: Sequence) -> None:
        logger.warning("preempting seq %s (recompute)", seq.request_id)
        self._release(seq)
        seq.status = SeqStatus.WAITING
        seq.num_preemptions += 1
        seq.prefilled = 0  # cache gone: the whole context re-prefills
        self.running.remove(seq)
        self.waiting.appendleft(seq)
