#!/usr/bin/env python3
"""Standalone decode-attention microbenchmark (GPU box tool, not a pytest).

Times paged_decode_attention at the serving shape (Tower-Plus-9B/gemma2:
B=256 seqs, ctx ~1064, 16 q heads / 8 kv heads, D=256, block 16) and
reports ms + achieved KV-read bandwidth. Run under rocprofv3 --pmc for
counter analysis.

  python tests/attn_microbench.py [--ctx 1064] [--batch 256] [--iters 50]
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from llmq_amd import ops  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--ctx", type=int, default=1064)
    ap.add_argument("--heads", type=int, default=16)
    ap.add_argument("--kv-heads", type=int, default=8)
    ap.add_argument("--head-dim", type=int, default=256)
    ap.add_argument("--block-size", type=int, default=16)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--softcap", type=float, default=0.0)
    ap.add_argument("--window", type=int, default=0)
    ap.add_argument("--check", action="store_true", help="verify vs torch ref")
    ap.add_argument("--fp8", action="store_true", help="fp8 KV caches")
    ap.add_argument("--prefill", action="store_true",
                    help="time varlen flash prefill instead of decode")
    ap.add_argument("--ab", action="store_true",
                    help="interleaved A/B of decode variants (LLMQ_DECODE_PIPE"
                    " 0/64/128) with per-variant numerics check")
    args = ap.parse_args()

    if args.prefill:
        return prefill_bench(args)
    if args.ab:
        return ab_bench(args)
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        assert ops.has_hip_ext()
    dev = torch.device("cuda:0" if use_gpu else "cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32
    B, L, H, KVH, D, bs = (args.batch, args.ctx, args.heads, args.kv_heads,
                           args.head_dim, args.block_size)
    blocks_per_seq = (L + bs - 1) // bs
    num_blocks = B * blocks_per_seq + 1

    torch.manual_seed(0)
    q = torch.randn(B, H, D, device=dev, dtype=dtype)
    k_cache = torch.randn(num_blocks, KVH, bs, D, device=dev, dtype=dtype)
    v_cache = torch.randn(num_blocks, KVH, bs, D, device=dev, dtype=dtype)
    if args.fp8:
        k_cache = k_cache.to(torch.float8_e4m3fn)
        v_cache = v_cache.to(torch.float8_e4m3fn)
    block_tables = torch.arange(
        1, 1 + B * blocks_per_seq, device=dev, dtype=torch.int32
    ).reshape(B, blocks_per_seq)
    context_lens = torch.full((B,), L, device=dev, dtype=torch.int32)
    scale = D ** -0.5
    out = q

    def run() -> None:
        nonlocal_out[0] = ops.paged_decode_attention(
            q, k_cache, v_cache, block_tables, context_lens,
            scale, args.softcap, args.window,
        )

    nonlocal_out = [None]

    for _ in range(args.warmup):
        run()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        run()
    if use_gpu:
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    out = nonlocal_out[0]

    ebytes = 1 if args.fp8 else 2
    kv_bytes = 2 * B * L * KVH * D * ebytes  # K+V read once per (b, kh)
    print(f"shape B={B} L={L} H={H} KVH={KVH} D={D} bs={bs} fp8={args.fp8}")
    print(f"{dt * 1e3:.3f} ms/iter   KV {kv_bytes / 2**30:.2f} GiB   "
          f"{kv_bytes / dt / 1e12:.2f} TB/s effective")

    if args.check:
        from llmq_amd.ops import torch_ref

        ref = torch_ref.paged_decode_attention(
            q.float(), k_cache.float(), v_cache.float(), block_tables,
            context_lens, scale, args.softcap, args.window,
        ).float()
        err = (out.float() - ref).abs().max().item()
        print(f"max|err| vs f32 ref: {err:.4e}")
        assert err < (0.2 if args.fp8 else 0.05)  # fp8 quantised cache


def ab_bench(args) -> None:
    """Within-probe interleaved A/B of the decode kernel variants (guide
    §5.4 rule 24: perf-Δ claims come from interleaved rounds in ONE
    process; cross-process noise on these boxes is ~±7%)."""
    assert torch.cuda.is_available() and ops.has_hip_ext()
    dev = torch.device("cuda:0")
    B, L, H, KVH, D, bs = (args.batch, args.ctx, args.heads, args.kv_heads,
                           args.head_dim, args.block_size)
    blocks_per_seq = (L + bs - 1) // bs
    num_blocks = B * blocks_per_seq + 1
    torch.manual_seed(0)
    q = torch.randn(B, H, D, device=dev, dtype=torch.bfloat16)
    k_cache = torch.randn(num_blocks, KVH, bs, D, device=dev, dtype=torch.bfloat16)
    v_cache = torch.randn(num_blocks, KVH, bs, D, device=dev, dtype=torch.bfloat16)
    block_tables = torch.arange(
        1, 1 + B * blocks_per_seq, device=dev, dtype=torch.int32
    ).reshape(B, blocks_per_seq)
    context_lens = torch.full((B,), L, device=dev, dtype=torch.int32)
    scale = D ** -0.5
    kv_bytes = 2 * B * L * KVH * D * 2

    variants = ["0", "32", "64", "128"]

    def run_once():
        return ops.paged_decode_attention(
            q, k_cache, v_cache, block_tables, context_lens,
            scale, args.softcap, args.window,
        )

    # numerics first
    from llmq_amd.ops import torch_ref
    ref = torch_ref.paged_decode_attention(
        q.float(), k_cache.float(), v_cache.float(), block_tables,
        context_lens, scale, args.softcap, args.window,
    ).float()
    for v in variants:
        os.environ["LLMQ_DECODE_PIPE"] = v
        err = (run_once().float() - ref).abs().max().item()
        print(f"variant pipe={v}: max|err| vs f32 ref = {err:.4e}")
        assert err < 0.05, f"variant {v} numerics FAIL"

    rounds = 6
    times = {v: [] for v in variants}
    for r in range(rounds):
        for v in variants:
            os.environ["LLMQ_DECODE_PIPE"] = v
            for _ in range(3):
                run_once()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                run_once()
            torch.cuda.synchronize()
            times[v].append((time.perf_counter() - t0) / args.iters)
    print(f"\nshape B={B} L={L} H={H} KVH={KVH} D={D} bs={bs}  "
          f"({rounds} interleaved rounds × {args.iters} iters)")
    for v in variants:
        ts = sorted(times[v])
        med, best = ts[len(ts) // 2], ts[0]
        print(f"  pipe={v:>3}: median {med * 1e3:.3f} ms  best {best * 1e3:.3f} ms"
              f"   median {kv_bytes / med / 1e12:.2f} TB/s  best {kv_bytes / best / 1e12:.2f} TB/s")


def prefill_bench(args) -> None:
    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if use_gpu else "cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32
    B, L, H, KVH, D = 8, args.ctx, args.heads, args.kv_heads, args.head_dim
    torch.manual_seed(0)
    T = B * L
    q = torch.randn(T, H, D, device=dev, dtype=dtype)
    k = torch.randn(T, KVH, D, device=dev, dtype=dtype)
    v = torch.randn(T, KVH, D, device=dev, dtype=dtype)
    cu = torch.arange(0, T + 1, L, dtype=torch.int32, device=dev)
    scale = D ** -0.5

    def run():
        return ops.varlen_prefill_attention(q, k, v, cu, L, scale)

    flops = 4.0 * B * H * D * (L * (L + 1) / 2)  # causal QK^T + PV
    print(f"prefill B={B} L={L} H={H} KVH={KVH} D={D}")
    if args.check:
        from llmq_amd.ops import torch_ref

        ref = torch_ref.varlen_prefill_attention(
            q.float().cpu(), k.float().cpu(), v.float().cpu(), cu.cpu(), scale)
        for variant in ("0", "1"):
            os.environ["LLMQ_PREFILL_PIPE"] = variant
            err = (run().float().cpu() - ref).abs().max().item()
            print(f"pipe={variant}: max|err|: {err:.4e}")
            assert err < 0.05
    variants = ["0", "1"] if args.ab else [os.environ.get("LLMQ_PREFILL_PIPE", "1")]
    rounds = 5 if args.ab else 1
    times = {vv: [] for vv in variants}
    for _ in range(rounds):
        for vv in variants:
            os.environ["LLMQ_PREFILL_PIPE"] = vv
            for _ in range(args.warmup):
                run()
            if use_gpu:
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                run()
            if use_gpu:
                torch.cuda.synchronize()
            times[vv].append((time.perf_counter() - t0) / args.iters)
    for vv in variants:
        dt = sorted(times[vv])[len(times[vv]) // 2]
        print(f"pipe={vv}: {dt * 1e3:.3f} ms/iter   {flops / dt / 1e12:.1f} TF/s")


if __name__ == "__main__":
    main()
