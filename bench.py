#!/usr/bin/env python3
"""Flagship serving benchmark (driver contract).

Measures the BASELINE.json headline: output tokens/sec for Tower-Plus-9B
(Gemma-2-9B architecture, random-init bf16, synthetic prompts) served by
N data-parallel engine workers, one rank per MI355X over RCCL.

A "step" is one continuous-batching decode iteration of the full resident
batch (the engine's serving step: scheduler → hipGraph decode forward →
sampling → bookkeeping). The engine is pre-filled with --batch sequences of
--prompt-len synthetic tokens (default batch 512 — 288 GB HBM3E holds far
more resident sequences than the reference's 750 default, and the larger
M amortises the per-step weight stream); after --warmup untimed steps, EXACTLY
--steps steps are timed between barrier+synchronize fences; the slowest
rank's wall time is used. value = N_ranks × batch × steps ÷ max_elapsed.

Launch (multi-GPU, by the driver):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=64)
    parser.add_argument("--warmup", type=int, default=16)
    parser.add_argument("--model", default="tower-plus-9b")
    parser.add_argument("--batch", type=int, default=512,
                        help="resident sequences per GPU (max_num_seqs)")
    parser.add_argument("--prompt-len", type=int, default=1024)
    parser.add_argument("--max-model-len", type=int, default=4096)
    parser.add_argument("--temperature", type=float, default=0.7)
    parser.add_argument("--eager", action="store_true", help="disable hipGraphs")
    parser.add_argument("--kv-dtype", default="auto", choices=["auto", "fp8"],
                        help="KV cache storage dtype (fp8 = optional mode, NOT the headline)")
    parser.add_argument("--profile-steps", type=int, default=0,
                        help="extra untimed steps after the timed region (rocprof)")
    parser.add_argument("--tp", type=int, default=0,
                        help="tensor-parallel degree: all ranks serve ONE "
                        "model sharded over RCCL/xGMI (BASELINE config #4, "
                        "e.g. --tp 8 --model tower-plus-72b under torchrun "
                        "--nproc-per-node 8). Default 0 = data-parallel.")
    args = parser.parse_args()

    import torch

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1

    use_gpu = torch.cuda.is_available()
    tp_mode = args.tp > 1
    if tp_mode and world != args.tp:
        raise SystemExit(
            f"--tp {args.tp} needs exactly that many ranks (WORLD_SIZE={world}); "
            "launch via torchrun --nproc-per-node N bench.py --tp N"
        )
    if distributed:
        import torch.distributed as dist

        if use_gpu:
            torch.cuda.set_device(local_rank)
        # "nccl" IS RCCL on ROCm; gloo covers the CPU test path.
        dist.init_process_group(backend="nccl" if use_gpu else "gloo")
        if tp_mode:
            from llmq_amd.parallel import TPGroup, set_tp_group

            set_tp_group(TPGroup(rank, world, None))

    device = f"cuda:{local_rank}" if use_gpu else "cpu"

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine
    from llmq_amd.engine.sampling_params import SamplingParams

    model = args.model
    batch = args.batch
    prompt_len = args.prompt_len
    if not use_gpu:
        # CPU fallback (contract: must run everywhere) — tiny shape.
        model, batch, prompt_len = "tiny-llama", 4, 32

    cfg = EngineConfig(
        model=model,
        max_num_seqs=batch,
        max_model_len=args.max_model_len if use_gpu else 256,
        max_prefill_tokens=16384,
        load_weights=False,
        fast_init=True,
        device=device,
        enforce_eager=args.eager,
        kv_cache_dtype=args.kv_dtype,
        hipgraph_max_batch=batch,
        # TP replicas must make IDENTICAL scheduler/sampler decisions —
        # same seed everywhere; DP ranks decorrelate instead.
        seed=1234 if tp_mode else 1234 + rank,
    )
    t_init = time.perf_counter()
    if tp_mode:
        engine = LLMEngine(cfg, tp_rank=rank, tp_size=world)
    else:
        engine = LLMEngine(cfg)
    if rank == 0:
        print(f"[bench] engine init {time.perf_counter() - t_init:.1f}s", file=sys.stderr)

    # --- load the resident batch (synthetic prompts, random token ids)
    import numpy as np

    rng = np.random.default_rng(42 if tp_mode else 42 + rank)
    vocab = engine.spec.vocab_size
    params = SamplingParams(
        temperature=args.temperature, max_tokens=args.max_model_len, ignore_eos=True
    )
    req_tag = "tp" if tp_mode else str(rank)
    for i in range(batch):
        ids = rng.integers(0, vocab, size=prompt_len).tolist()
        engine.add_request(f"bench-{req_tag}-{i}", prompt_token_ids=ids, params=params)

    t_prefill = time.perf_counter()
    stalled = 0
    while engine.scheduler.num_waiting > 0 or engine.scheduler.prefilling:
        before = sum(sq.num_tokens for sq in engine.scheduler.running)
        engine.step()  # admission steps (prefill + ride-along decode)
        after = sum(sq.num_tokens for sq in engine.scheduler.running)
        # Fail loudly instead of spinning silently if admission makes no
        # progress (e.g. the KV pool can never seat the requested batch
        # and preemption churns running -> waiting forever).
        stalled = stalled + 1 if after <= before and engine.scheduler.num_waiting > 0 else 0
        if stalled >= 50:
            raise RuntimeError(
                f"admission stalled: {engine.scheduler.num_waiting} waiting, "
                f"{len(engine.scheduler.running)} running, {after} resident tokens "
                f"— KV pool too small for --batch x --max-model-len?"
            )
    prefill_s = time.perf_counter() - t_prefill
    prefill_tokens = batch * prompt_len
    # mixed steps decode already-admitted seqs while later ones prefill, so
    # the phase also produced decode tokens — count both for the rate.
    decoded = sum(sq.num_tokens for sq in engine.scheduler.running) - prefill_tokens
    if rank == 0:
        print(
            f"[bench] admission phase: {prefill_tokens} prefill + {decoded} decode "
            f"tokens in {prefill_s:.2f}s "
            f"({(prefill_tokens + decoded) / prefill_s:.0f} tok/s/gpu)",
            file=sys.stderr,
        )

    def barrier_sync():
        if use_gpu:
            torch.cuda.synchronize()
        if distributed:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    # --- warmup
    for _ in range(args.warmup):
        engine.step()
    barrier_sync()

    # --- timed region
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.step()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    barrier_sync()

    # slowest rank defines the job time
    if distributed:
        t = torch.tensor([elapsed], device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    assert engine.scheduler.num_running == batch, (
        f"batch decayed during timing: {engine.scheduler.num_running} != {batch}"
    )

    # TP shards ONE model over all ranks (whole-job tokens = one engine's);
    # DP runs an independent engine per rank.
    total_tokens = (1 if tp_mode else world) * batch * args.steps
    value = total_tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    for _ in range(args.profile_steps):
        engine.step()
    if use_gpu:
        torch.cuda.synchronize()

    if rank == 0:
        result = {
            "metric": (
                "output_tokens_per_sec "
                f"({'Tower-Plus-9B' if args.model == 'tower-plus-9b' else args.model}"
                " continuous-batching serving decode)"
            ),
            "value": round(value, 1),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": (
                ("bf16" if args.kv_dtype == "auto" else "bf16+fp8kv")
                if use_gpu else "fp32"
            ),
            "data": "synthetic",
            "config": {
                "model": f"{model} ({engine.spec.name} arch, random init)",
                "global_batch": batch if tp_mode else world * batch,
                "seq_len": prompt_len,
                "parallelism": f"tp{world}" if tp_mode else f"dp{world}",
                "kv_cache_dtype": args.kv_dtype,
            },
        }
        # completed jobs/sec companion number: at steady-state decode one
        # job finishes per output_len generated tokens.
        out_len = args.max_model_len - prompt_len if use_gpu else 8
        result["config"]["jobs_per_sec_at_output_len"] = {
            str(out_len): round(value / max(out_len, 1), 3)
        }
        print(json.dumps(result))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
