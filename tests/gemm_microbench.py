#!/usr/bin/env python3
"""Skinny-GEMM microbenchmark (GPU box tool): ops.skinny_gemm vs hipBLASLt
(F.linear) at the exact decode-projection shapes of the serving step.

  python tests/gemm_microbench.py [--m 512] [--check] [--iters 50]
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from llmq_amd import ops  # noqa: E402

# gemma-2-9b serving shapes: (name, K, N, bias)
SHAPES = [
    ("qkv", 3584, 8192, False),
    ("o", 4096, 3584, False),
    ("gate_up", 3584, 28672, False),
    ("down", 14336, 3584, False),
    ("logits", 3584, 256128, False),
]


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=512)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--splitk", type=int, default=0, help="0 = auto")
    args = ap.parse_args()
    assert torch.cuda.is_available() and ops.has_hip_ext()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    M = args.m

    total_ours = total_blas = 0.0
    for name, K, N, has_bias in SHAPES:
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        w = (torch.randn(N, K, device=dev) * 0.02).bfloat16()
        bias = (torch.randn(N, device=dev) * 0.1).bfloat16() if has_bias else None
        out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
        flops = 2.0 * M * K * N

        if args.check:
            ops.skinny_gemm(x, w, bias, splitk=args.splitk, out=out)
            ref = torch.nn.functional.linear(x.float(), w.float(),
                                             bias.float() if bias is not None else None)
            err = (out.float() - ref).abs()
            rel = (err / ref.abs().clamp_min(1.0)).max().item()
            print(f"{name}: max|err|={err.max().item():.4f} rel={rel:.5f}")
            assert rel < 0.02, f"{name} numerics FAIL"

        def bench(fn):
            for _ in range(args.warmup):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                fn()
            torch.cuda.synchronize()
            return (time.perf_counter() - t0) / args.iters

        t_ours = bench(lambda: ops.skinny_gemm(x, w, bias, splitk=args.splitk, out=out))
        t_blas = bench(lambda: torch.nn.functional.linear(x, w, bias))
        total_ours += t_ours
        total_blas += t_blas
        print(f"{name:8s} M={M} K={K:6d} N={N:6d}: "
              f"ours {t_ours * 1e6:8.1f} us ({flops / t_ours / 1e12:7.1f} TF)  "
              f"blas {t_blas * 1e6:8.1f} us ({flops / t_blas / 1e12:7.1f} TF)  "
              f"ratio {t_blas / t_ours:.2f}x")
    print(f"TOTAL: ours {total_ours * 1e3:.3f} ms   blas {total_blas * 1e3:.3f} ms  "
          f"({total_blas / total_ours:.2f}x)")


if __name__ == "__main__":
    main()
