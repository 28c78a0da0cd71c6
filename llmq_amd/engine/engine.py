"""LLMEngine — synchronous continuous-batching engine.

One ``step()`` runs one scheduler iteration (a prefill batch or a decode
batch), samples, advances sequences, and returns per-request progress.
The async facade (workers/engine_worker.py AsyncEngineBridge) drives this
loop on a dedicated engine thread.

This is the MI355X-native replacement for the vLLM AsyncLLMEngine the
reference consumes (vllm_worker.py:105-123 construction, 183-186
generate stream): continuous batching up to max_num_seqs,
paged KV sized for 288 GB HBM3E, hipGraph decode steps, TP over RCCL.
"""

from __future__ import annotations

import logging
import os
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from llmq_amd import ops
from llmq_amd.engine.config import EngineConfig
from llmq_amd.engine.kv_cache import BlockAllocator, KVCache
from llmq_amd.engine.model_runner import ModelRunner
from llmq_amd.engine.model_specs import ModelSpec, resolve_spec
from llmq_amd.engine.models.llama import CausalLM
from llmq_amd.engine.sampling_params import SamplingParams
from llmq_amd.engine.scheduler import Scheduler, Sequence
from llmq_amd.engine.tokenizer import load_tokenizer

logger = logging.getLogger(__name__)

# Memory the engine leaves free on top of weights+KV (workspace, graphs).
_RESERVE_BYTES = 4 << 30

# Mixed-step policy: fusing decode+prefill into one forward (shared GEMMs)
# measured WORSE than splitting at every admission size on MI355X
# (profiles/r2_step2: steady-state 3,967 fused vs 4,247 split tok/s) — the
# decode segment loses its hipGraph and the fused eager forward costs more
# than the extra weight read. Default: ALWAYS split (replay the decode
# graph + run the prefill segment separately, overlapped on two streams).
# Env-overridable for A/B: a small value re-enables fusion above that many
# prefill tokens.
_MIXED_FUSE_MIN_TOKENS = int(
    os.environ.get("LLMQ_MIXED_FUSE_MIN_TOKENS", str(1 << 30)))

_TUNED_GEMMS_DONE = False


def _enable_tuned_gemms() -> None:
    """Load the shipped TunableOp results (hipBLASLt/rocBLAS algorithm
    selections tuned on MI355X for the serving GEMM shapes). No-op when the
    user drives TunableOp via PYTORCH_TUNABLEOP_* env vars themselves."""
    global _TUNED_GEMMS_DONE
    if _TUNED_GEMMS_DONE or os.environ.get("PYTORCH_TUNABLEOP_ENABLED"):
        return
    _TUNED_GEMMS_DONE = True
    from pathlib import Path

    csv = Path(__file__).parent.parent / "ops" / "tunableop_gfx950.csv"
    if not csv.is_file():
        return
    try:
        import torch.cuda.tunable as tunable

        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(str(csv))
        logger.info("TunableOp GEMM selections loaded from %s", csv.name)
    except Exception as exc:  # noqa: BLE001 — tuning is an optimisation only
        logger.warning("TunableOp setup failed (%s); using default GEMMs", exc)


@dataclass(slots=True)
class RequestOutput:
    request_id: str
    new_token_ids: List[int] = field(default_factory=list)
    finished: bool = False
    finish_reason: Optional[str] = None
    text: str = ""
    prompt_tokens: int = 0
    output_tokens: int = 0
    queue_wait_ms: Optional[float] = None
    prefill_ms: Optional[float] = None
    decode_ms: Optional[float] = None


class LLMEngine:
    def __init__(self, config: EngineConfig, tp_rank: int = 0, tp_size: Optional[int] = None):
        self.config = config
        self.device = config.resolve_device()
        self.dtype = config.resolve_dtype(self.device)
        self.spec: ModelSpec = resolve_spec(config.model)
        tp = tp_size if tp_size is not None else config.tensor_parallel_size
        self.tp_size = tp
        self.tp_rank = tp_rank
        if self.device.type == "cuda" and not ops.has_hip_ext():
            # Fail loudly: GPU execution must run the CDNA4 kernels.
            ops._use_hip(torch.empty(1, device=self.device))
        if self.device.type == "cuda":
            _enable_tuned_gemms()
        self.max_model_len = min(
            config.max_model_len or self.spec.max_position_embeddings,
            self.spec.max_position_embeddings,
        )
        logger.info(
            "initialising engine: model=%s (%s, %.2fB params) device=%s dtype=%s tp=%d",
            self.spec.name, self.spec.family, self.spec.param_count() / 1e9,
            self.device, self.dtype, tp,
        )
        self.model = CausalLM(self.spec, self.device, self.dtype, tp_rank, tp)
        self._load_or_init_weights()
        self.tokenizer = load_tokenizer(
            config.model, self.spec.vocab_size, self.spec.bos_token_id, self.spec.eos_token_id
        )
        self.kv_dtype = config.resolve_kv_dtype(self.device)
        if self.kv_dtype == torch.float8_e4m3fn and self.spec.head_dim not in (128, 256):
            raise ValueError("fp8 KV cache requires head_dim 128 or 256 (MFMA decode path)")
        num_blocks = self._size_kv_cache()
        if tp > 1:
            # Deterministic lockstep requires an IDENTICAL KV pool on every
            # rank: mem_get_info varies per GPU, and different block counts
            # mean divergent preemption/admission → silent token divergence
            # between TP replicas. Take the min so every rank fits.
            from llmq_amd.parallel import get_tp_group

            synced = get_tp_group().min_scalar(num_blocks)
            if synced != num_blocks:
                logger.info(
                    "KV blocks synced across TP ranks: local %d -> min %d",
                    num_blocks, synced,
                )
            num_blocks = synced
        kv_heads_local = self.spec.num_kv_heads // tp
        self.kv_cache = KVCache(
            self.spec.num_layers, num_blocks, kv_heads_local,
            config.kv_block_size, self.spec.head_dim, self.device, self.kv_dtype,
        )
        self.allocator = BlockAllocator(num_blocks)
        self.scheduler = Scheduler(
            self.allocator,
            config.kv_block_size,
            config.max_num_seqs,
            config.max_prefill_tokens,
            self.max_model_len,
        )
        self.runner = ModelRunner(
            self.model, self.kv_cache, config, self.device, self.max_model_len
        )
        # Mixed-step overlap: decode graph on a side stream ∥ prefill on the
        # default stream (disjoint KV slots; separate sampler buffers).
        # gemma-2-27b deadlocks the GPU when its decode and prefill work run
        # concurrently on two streams: GPU busy 100 %, zero memory traffic.
        # ROOT CAUSE (bisected on MI355X, profiles/r2_step5): its GEMM
        # shapes select hipBLASLt stream-k kernels (SK3) whose workgroups
        # spin on global tile counters assuming the whole grid is resident;
        # with a second stream's kernels holding CUs, part of the stream-k
        # grid never launches and the resident part spins forever. Proof:
        # TENSILE_STREAMK_DATA_PARALLEL=1 (no cross-WG sync) unhangs the
        # overlap — but also slows 27b's decode GEMMs 37 %, so overlap-off
        # (sequential split steps) is both the SAFE and the FASTEST config
        # for this model. LLMQ_OVERLAP_MIXED=1 forces the overlap back on
        # (combine with TENSILE_STREAMK_DATA_PARALLEL=1 to avoid the hang).
        overlap_env = os.environ.get("LLMQ_OVERLAP_MIXED")
        overlap_default = not (
            self.spec.family == "gemma2" and self.spec.head_dim == 128
        )
        self._overlap_mixed = (
            self.device.type == "cuda"
            and tp == 1
            and (
                overlap_env not in ("0", "false")
                if overlap_env is not None
                else overlap_default
            )
        )
        self._decode_stream = (
            torch.cuda.Stream(device=self.device) if self._overlap_mixed else None
        )
        # diagnostics: run the overlap's decode EAGER (no graph replay), or
        # serialise the two streams (two-stream plumbing, zero concurrency)
        self._overlap_eager = os.environ.get("LLMQ_OVERLAP_EAGER") == "1"
        self._overlap_sync = os.environ.get("LLMQ_OVERLAP_SYNC") == "1"
        self.runner.capture_graphs()
        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            if free < (12 << 30):
                # Prefill activations at max_prefill_tokens rows (plus
                # hipBLASLt workspaces) live OUTSIDE the KV budget; with
                # single-digit GB of headroom the allocator thrashes into
                # retry storms (observed with 72B: 144 GB weights + a full
                # KV pool + 16k-row prefill segments). Lower
                # gpu_memory_utilization, max_model_len·max_num_seqs (the
                # KV cap) or max_prefill_tokens.
                logger.warning(
                    "only %.1f GB of HBM headroom left after weights+KV+"
                    "graphs — large prefill segments may thrash the "
                    "allocator (reduce max_prefill_tokens=%d or the KV cap)",
                    free / 2**30, config.max_prefill_tokens,
                )
        self._seqs: Dict[str, Sequence] = {}
        self._prefill_done_at: Dict[str, float] = {}
        self.steps = 0
        logger.info(
            "engine ready: %d KV blocks (%d tokens, %.1f GB), max_num_seqs=%d, graphs=%s",
            num_blocks, num_blocks * config.kv_block_size,
            num_blocks * KVCache.block_bytes(
                self.spec.num_layers, kv_heads_local, config.kv_block_size,
                self.spec.head_dim, self.kv_dtype,
            ) / 2 ** 30 * 1.0,
            config.max_num_seqs,
            self.runner.use_graphs,
        )

    # -- init helpers ----------------------------------------------------

    def _load_or_init_weights(self) -> None:
        from pathlib import Path

        path = Path(self.config.model)
        if self.config.load_weights and path.is_dir() and any(path.glob("*.safetensors")):
            from llmq_amd.engine.weights import load_safetensors_weights

            load_safetensors_weights(self.model, path)
        else:
            self.model.random_init(self.config.seed, fast=self.config.fast_init)

    def _size_kv_cache(self) -> int:
        cfg = self.config
        kv_heads_local = self.spec.num_kv_heads // self.tp_size
        if cfg.num_kv_blocks is not None:
            return cfg.num_kv_blocks
        if self.device.type != "cuda":
            # CPU tests: enough for max_num_seqs × a modest context.
            return max(
                256,
                (cfg.max_num_seqs * min(self.max_model_len, 1024)) // cfg.kv_block_size,
            )
        free, total = torch.cuda.mem_get_info(self.device)
        used = total - free
        budget = int(total * cfg.gpu_memory_utilization) - used - _RESERVE_BYTES
        num = KVCache.num_blocks_for_budget(
            budget, self.spec.num_layers, kv_heads_local, cfg.kv_block_size,
            self.spec.head_dim, self.kv_dtype,
        )
        # No point holding more KV than every admitted seq at full context.
        cap = (cfg.max_num_seqs * self.max_model_len) // cfg.kv_block_size + cfg.max_num_seqs
        num = min(num, cap)
        if num < 16:
            raise RuntimeError(
                f"KV budget too small: {budget / 2**30:.1f} GB free for KV "
                f"(gpu_memory_utilization={cfg.gpu_memory_utilization}; the "
                "utilization caps TOTAL device usage incl. other processes — "
                "when sharing a GPU, raise it and bound KV via max_model_len/"
                "max_num_seqs instead)"
            )
        return num

    # -- request API -----------------------------------------------------

    def add_request(
        self,
        request_id: str,
        prompt: Optional[str] = None,
        prompt_token_ids: Optional[List[int]] = None,
        params: Optional[SamplingParams] = None,
    ) -> None:
        if request_id in self._seqs:
            raise ValueError(f"duplicate request_id {request_id}")
        params = params or SamplingParams()
        if prompt_token_ids is None:
            if prompt is None:
                raise ValueError("need prompt or prompt_token_ids")
            prompt_token_ids = self.tokenizer.encode(prompt)
        if not prompt_token_ids:
            prompt_token_ids = [self.spec.bos_token_id]
        seq = Sequence(request_id, prompt_token_ids, params, arrival_time=time.time())
        self._seqs[request_id] = seq
        self.scheduler.add(seq)

    def abort_request(self, request_id: str) -> None:
        self.scheduler.abort(request_id)
        self._seqs.pop(request_id, None)
        self._prefill_done_at.pop(request_id, None)

    def has_unfinished(self) -> bool:
        return self.scheduler.has_work()

    def num_unfinished(self) -> int:
        return self.scheduler.num_waiting + self.scheduler.num_running

    # -- stepping --------------------------------------------------------

    last_step_kind: str = "decode"  # observability: kind of the last step

    def step(self) -> List[RequestOutput]:
        batch = self.scheduler.schedule()
        if batch.empty:
            return []
        self.last_step_kind = batch.kind
        t0 = time.perf_counter()
        if batch.kind == "prefill":
            tokens = self.runner.execute_prefill(batch.seqs, batch.chunks or None)
        elif batch.kind == "mixed":
            ptoks = sum(e - st for st, e in batch.chunks)
            if ptoks < _MIXED_FUSE_MIN_TOKENS and self.runner.use_graphs:
                # SPLIT policy (default, measured fastest): replay the
                # decode hipGraph and run the prefill segment as its own
                # forward. On one GPU the two are OVERLAPPED on separate
                # HIP streams — decode attention is HBM-bound while prefill
                # GEMMs are MFMA-bound, so the hardware runs them
                # concurrently (profiles/r2_step2). TP replicas stay
                # sequential: stream interleaving would make rank-local
                # RNG/collective ordering nondeterministic.
                dseqs = batch.seqs[: batch.n_decode]
                pseqs = batch.seqs[batch.n_decode :]
                if self._overlap_mixed:
                    ds = self._decode_stream
                    cur = torch.cuda.current_stream()
                    ds.wait_stream(cur)
                    with torch.cuda.stream(ds):
                        # .clone(): the fused sampler returns a VIEW of a
                        # persistent output buffer
                        d_tokens = self.runner.execute_decode(
                            dseqs, force_eager=self._overlap_eager).clone()
                    if self._overlap_sync:
                        ds.synchronize()
                    p_tokens = self.runner.execute_prefill(
                        pseqs, batch.chunks, buf_name="prefill"
                    )
                    cur.wait_stream(ds)
                    d_tokens.record_stream(cur)
                    tokens = torch.cat([d_tokens, p_tokens])
                else:
                    d_tokens = self.runner.execute_decode(dseqs).clone()
                    p_tokens = self.runner.execute_prefill(pseqs, batch.chunks)
                    tokens = torch.cat([d_tokens, p_tokens])
            else:
                tokens = self.runner.execute_mixed(batch)
        else:
            tokens = self.runner.execute_decode(batch.seqs)
        token_list = tokens.tolist()
        now = time.time()
        self.steps += 1
        outputs: List[RequestOutput] = []
        # Only rows that SAMPLED get a token: every decode row, plus prefill
        # rows whose context completed this step (mid-chunks of long prompts
        # produce no token yet — chunked prefill).
        if batch.kind == "decode":
            sampled_seqs = batch.seqs
            first_prefill = len(batch.seqs)
        else:
            pseqs = batch.seqs[batch.n_decode:]
            chunks = batch.chunks or [(0, sq.num_tokens) for sq in pseqs]
            final = []
            for sq, (st, e) in zip(pseqs, chunks):
                if st == 0:
                    sq.prefill_start_time = now
                sq.prefilled = e
                if e == sq.num_tokens:
                    final.append(sq)
            sampled_seqs = batch.seqs[: batch.n_decode] + final
            first_prefill = batch.n_decode
        eos_id = self.tokenizer.eos_token_id
        mml = self.max_model_len
        for row, (seq, tok) in enumerate(zip(sampled_seqs, token_list)):
            if row >= first_prefill:
                self._prefill_done_at[seq.request_id] = now
                if seq.first_token_time is None:
                    seq.first_token_time = now
            seq.token_ids.append(tok)
            params = seq.params
            nt = len(seq.token_ids)
            out = RequestOutput(
                request_id=seq.request_id,
                new_token_ids=[tok],
                prompt_tokens=seq.prompt_len,
                output_tokens=nt - seq.prompt_len,
            )
            # Cheap pre-check: most decode steps finish nothing; only take
            # the full _check_finish path (stop-string decode etc.) when a
            # finish condition can actually hold.
            maybe_done = (
                (tok == eos_id and not params.ignore_eos)
                or nt - seq.prompt_len >= params.max_tokens
                or nt >= mml
                or params.stop
            )
            reason = self._check_finish(seq) if maybe_done else None
            if reason is not None:
                self.scheduler.finish(seq, reason)
                out.finished = True
                out.finish_reason = reason
                out.text = self._final_text(seq)
                prefill_at = self._prefill_done_at.pop(seq.request_id, now)
                started = seq.prefill_start_time or prefill_at
                out.queue_wait_ms = (started - seq.arrival_time) * 1000.0
                out.prefill_ms = (prefill_at - started) * 1000.0
                out.decode_ms = (now - prefill_at) * 1000.0
                self._seqs.pop(seq.request_id, None)
            outputs.append(out)
        return outputs

    def _check_finish(self, seq: Sequence) -> Optional[str]:
        params = seq.params
        last = seq.token_ids[-1]
        if not params.ignore_eos and last == self.tokenizer.eos_token_id:
            return "eos"
        if seq.output_len >= params.max_tokens:
            return "length"
        if seq.num_tokens >= self.max_model_len:
            return "length"
        if params.stop:
            tail_ids = seq.token_ids[seq.prompt_len:]
            text = self.tokenizer.decode(tail_ids)
            for stop in params.stop:
                idx = text.find(stop)
                if idx >= 0:
                    seq.output_text = text[:idx]
                    return "stop"
        return None

    def _final_text(self, seq: Sequence) -> str:
        if seq.output_text is not None:
            return seq.output_text
        out_ids = seq.token_ids[seq.prompt_len:]
        if out_ids and out_ids[-1] == self.tokenizer.eos_token_id:
            out_ids = out_ids[:-1]
        return self.tokenizer.decode(out_ids)

    # -- convenience (tests / bench) -------------------------------------

    def generate_batch(
        self, prompts: List[str], params: Optional[SamplingParams] = None
    ) -> List[str]:
        """Blocking helper: run all prompts to completion, return texts."""
        results: Dict[str, str] = {}
        for i, p in enumerate(prompts):
            self.add_request(f"batch-{i}", prompt=p, params=params)
        while self.has_unfinished():
            for out in self.step():
                if out.finished:
                    results[out.request_id] = out.text
        return [results[f"batch-{i}"] for i in range(len(prompts))]
