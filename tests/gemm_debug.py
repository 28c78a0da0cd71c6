#!/usr/bin/env python3
"""Isolate the skinny_gemm split-K NaN (GPU box tool)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from llmq_amd import ops

dev = torch.device("cuda:0")
torch.manual_seed(0)
M, K, N = 512, 3584, 8192
x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
w = (torch.randn(N, K, device=dev) * 0.02).bfloat16()
ref = torch.nn.functional.linear(x.float(), w.float())
for z in (1, 2, 4):
    out = torch.full((M, N), float("nan"), dtype=torch.bfloat16, device=dev)
    ops.skinny_gemm(x, w, None, splitk=z, out=out)
    err = (out.float() - ref).abs()
    nan_rows = torch.isnan(out.float()).any(dim=1).sum().item()
    nan_cols = torch.isnan(out.float()).any(dim=0).sum().item()
    nan_total = torch.isnan(out.float()).sum().item()
    print(f"z={z}: max|err|={err.nan_to_num(1e9).max().item():.5f} "
          f"nan_elems={nan_total} nan_rows={nan_rows} nan_cols={nan_cols}")
    if nan_total:
        idx = torch.isnan(out.float()).nonzero()[:8]
        print("  first nan positions:", idx.tolist())
# also M=256
M = 256
x2 = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
ref2 = torch.nn.functional.linear(x2.float(), w.float())
for z in (1, 2, 4):
    out = torch.full((M, N), float("nan"), dtype=torch.bfloat16, device=dev)
    ops.skinny_gemm(x2, w, None, splitk=z, out=out)
    err = (out.float() - ref2).abs()
    print(f"M=256 z={z}: max|err|={err.nan_to_num(1e9).max().item():.5f} "
          f"nans={torch.isnan(out.float()).sum().item()}")
