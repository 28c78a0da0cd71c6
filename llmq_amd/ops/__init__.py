"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, torch reference on CPU.

Policy (per the build contract): on a GPU box the HIP extension MUST be
present and is ALWAYS used — if `torch.cuda.is_available()` and the
extension failed to load, every op raises instead of silently falling back
to eager PyTorch. On CPU (tests, this dev container) the torch reference
implementations run.
"""

from __future__ import annotations

import os
from pathlib import Path
from typing import Optional, Tuple

import torch

from llmq_amd.ops import torch_ref

_EXT = None
_EXT_ERROR: Optional[str] = None
_EXT_TRIED = False


def _try_load_extension() -> None:
    """Lazy: called on first use, NOT at import (so `python -m
    llmq_amd.ops.build` can rebuild without a namespace clash against a
    stale loaded .so)."""
    global _EXT, _EXT_ERROR, _EXT_TRIED
    if _EXT_TRIED:
        return
    _EXT_TRIED = True
    here = Path(__file__).parent
    override = os.environ.get("LLMQ_OPS_SO")
    if override:
        candidates = [Path(override)]
    else:
        candidates = sorted(here.glob("_hip_ops*.so"))
    if not candidates:
        _EXT_ERROR = (
            f"HIP extension not built (no _hip_ops*.so under {here}). "
            "Run `python -m llmq_amd.ops.build` (or __graft_entry__.build())."
        )
        return
    try:
        torch.ops.load_library(str(candidates[0]))
        _EXT = torch.ops.llmq_amd
        _EXT_ERROR = None
    except Exception as exc:  # noqa: BLE001
        # If an identical build was already loaded in this process (e.g.
        # right after `build()`), the namespace exists — use it.
        try:
            torch.ops.llmq_amd.rmsnorm  # noqa: B018
            _EXT = torch.ops.llmq_amd
            _EXT_ERROR = None
        except Exception:
            _EXT_ERROR = f"failed to load {candidates[0]}: {exc}"


def has_hip_ext() -> bool:
    _try_load_extension()
    return _EXT is not None


def _use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    _try_load_extension()
    if _EXT is None:
        raise RuntimeError(
            f"llmq_amd HIP extension required for GPU execution but unavailable: {_EXT_ERROR}"
        )
    return True


# ---------------------------------------------------------------- norms --


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float, offset: float = 0.0) -> torch.Tensor:
    if _use_hip(x):
        out = torch.empty_like(x)
        _EXT.rmsnorm(out, x, weight, eps, offset)
        return out
    return torch_ref.rmsnorm(x, weight, eps, offset)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float, offset: float = 0.0
) -> Tuple[torch.Tensor, torch.Tensor]:
    """In place on GPU: residual += x, x = rmsnorm(residual). Returns (x, residual)."""
    if _use_hip(x):
        _EXT.fused_add_rmsnorm(x, residual, weight, eps, offset)
        return x, residual
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps, offset)


def norm_add_norm(
    x: torch.Tensor, residual: torch.Tensor, w_post: torch.Tensor,
    w_pre: torch.Tensor, eps: float, offset: float = 0.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Gemma-2 sandwich: residual += rmsnorm(x, w_post);
    x = rmsnorm(residual, w_pre). In place on GPU; returns (x, residual)."""
    if _use_hip(x):
        _EXT.norm_add_norm(x, residual, w_post, w_pre, eps, offset)
        return x, residual
    return torch_ref.norm_add_norm(x, residual, w_post, w_pre, eps, offset)


# ----------------------------------------------------------------- rope --


def rope_inplace(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor
) -> None:
    if _use_hip(q):
        _EXT.rope_inplace(q, k, positions, cos_sin)
        return
    torch_ref.rope_inplace(q, k, positions, cos_sin)


build_rope_cache = torch_ref.build_rope_cache


# ----------------------------------------------------------- activations --


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if _use_hip(x):
        d = x.shape[-1] // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        _EXT.silu_and_mul(out, x)
        return out
    return torch_ref.silu_and_mul(x)


def gelu_tanh_and_mul(x: torch.Tensor) -> torch.Tensor:
    if _use_hip(x):
        d = x.shape[-1] // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        _EXT.gelu_tanh_and_mul(out, x)
        return out
    return torch_ref.gelu_tanh_and_mul(x)


def rope_and_cache(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    positions: torch.Tensor, cos_sin: torch.Tensor, slot_mapping: torch.Tensor,
) -> None:
    """Fused: RoPE(q, k) in place + scatter rotated k and v into the paged
    cache (one launch instead of two per layer)."""
    if _use_hip(q):
        _EXT.rope_and_cache(q, k, v, k_cache, v_cache, positions, cos_sin, slot_mapping)
        return
    torch_ref.rope_inplace(q, k, positions, cos_sin)
    torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


# ------------------------------------------------------------- KV cache --


def reshape_and_cache(
    key: torch.Tensor,
    value: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if _use_hip(key):
        _EXT.reshape_and_cache(key, value, k_cache, v_cache, slot_mapping)
        return
    torch_ref.reshape_and_cache(key, value, k_cache, v_cache, slot_mapping)


# ------------------------------------------------------------ attention --


def paged_decode_attention(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    softcap: float = 0.0,
    window: int = 0,
) -> torch.Tensor:
    if _use_hip(q):
        out = torch.empty_like(q)
        _EXT.paged_decode_attention(
            out, q, k_cache, v_cache, block_tables, context_lens, scale, softcap, window
        )
        return out
    return torch_ref.paged_decode_attention(
        q, k_cache, v_cache, block_tables, context_lens, scale, softcap, window
    )


def varlen_prefill_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    max_seqlen: int,
    scale: float,
    softcap: float = 0.0,
    window: int = 0,
    cu_seqlens_k: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Packed causal attention; cu_seqlens_k (key offsets per seq, >= query
    lengths) enables chunked prefill: Q is the last Lq positions of a
    Lk-long context."""
    cu_k = cu_seqlens if cu_seqlens_k is None else cu_seqlens_k
    if _use_hip(q):
        out = torch.empty_like(q)
        _EXT.varlen_prefill_attention(
            out, q, k, v, cu_seqlens, cu_k, int(max_seqlen), scale, softcap, window
        )
        return out
    return torch_ref.varlen_prefill_attention(
        q, k, v, cu_seqlens, scale, softcap, window, cu_seqlens_k=cu_k
    )


# ---------------------------------------------------------- skinny GEMM --


def skinny_gemm(
    x: torch.Tensor,          # [M, K] bf16
    w: torch.Tensor,          # [N, K] bf16 (torch linear weight layout)
    bias: Optional[torch.Tensor] = None,
    splitk: int = 0,          # 0 = auto
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Hand-written skinny-M decode-projection GEMM (out = x @ w.T + bias).

    256x256x64 MFMA tile, glds-staged with source-side st_16x32 swizzle,
    counted-vmcnt pipeline; split-K slabs + fused reduce for narrow-N
    shapes. Falls back to F.linear off-GPU / unsupported shapes."""
    M, K = x.shape
    N = w.shape[0]
    if not (x.is_cuda and x.dtype == torch.bfloat16 and K % 64 == 0
            and N % 4 == 0 and has_hip_ext()):
        import torch.nn.functional as F

        return F.linear(x, w, bias)
    if out is None:
        out = torch.empty(M, N, dtype=torch.bfloat16, device=x.device)
    _EXT.skinny_gemm(out, x, w, bias, splitk)
    return out


# ------------------------------------------------------------- sampling --


def sample_gumbel_argmax(
    out: torch.Tensor,        # [B] int64
    keys: torch.Tensor,       # [B] int64 scratch (packed value|~index)
    logits: torch.Tensor,     # [B, V] fp32
    temps: torch.Tensor,      # [B] fp32 (<= 0 → greedy row)
    req_seeds: torch.Tensor,  # [B] int32 (0 = unseeded)
    req_pos: torch.Tensor,    # [B] int32 output position (seeded rows)
    seed: int,
    step: int,
    bounds: Optional[torch.Tensor] = None,  # [B] fp32 top-k/p keep-bound
) -> None:
    """Fused one-pass sampler: per-row Gumbel-max (== softmax sampling) or
    argmax for greedy rows, optionally truncated to {logit >= bounds[b]}
    (top-k/top-p, from topk_topp_bound). Unseeded rows draw from the
    counter-based (engine seed, step, row) stream; rows with a request seed
    draw from (request_seed, output_position) — reproducible for a given
    request regardless of batch placement."""
    assert _use_hip(logits)
    keys.zero_()
    _EXT.sample_gumbel_argmax(out, keys, logits, temps, req_seeds, req_pos,
                              seed, step, bounds)


def topk_topp_bound(
    logits: torch.Tensor,   # [B, V] fp32
    temps: torch.Tensor,    # [B] fp32
    top_ps: torch.Tensor,   # [B] fp32
    top_ks: torch.Tensor,   # [B] int64 (0 = off)
) -> torch.Tensor:
    """Per-row logit-space keep-bound realising top-p ∧ top-k truncation
    via histogram select (3 streaming passes; no full-vocab sort)."""
    assert _use_hip(logits)
    out = torch.empty(logits.size(0), dtype=torch.float32, device=logits.device)
    _EXT.topk_topp_bound(out, logits, temps, top_ps, top_ks)
    return out


def sample_tokens(
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_ps: torch.Tensor,
    top_ks: torch.Tensor,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    # Generic path (top-k/top-p capable). The serving hot loop uses the
    # fused sample_gumbel_argmax above when no top-k/top-p is requested.
    if logits.is_cuda:
        return _sample_tokens_gpu(logits, temperatures, top_ps, top_ks, generator)
    return torch_ref.sample_tokens(logits, temperatures, top_ps, top_ks, generator)


def _sample_tokens_gpu(
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_ps: torch.Tensor,
    top_ks: torch.Tensor,
    generator: Optional[torch.Generator],
) -> torch.Tensor:
    """Batched GPU sampling without per-sequence python loops."""
    B, V = logits.shape
    greedy = temperatures <= 0
    temps = torch.where(greedy, torch.ones_like(temperatures), temperatures)
    scaled = logits.float() / temps.unsqueeze(1)
    need_topk = bool((top_ks > 0).any()) and bool((top_ks < V).any())
    need_topp = bool((top_ps < 1.0).any())
    if need_topk or need_topp:
        # histogram-select keep-bound (3 streaming passes) instead of the
        # full-vocab sort — at B=256 × V=256k the sort was a multi-ms cliff
        bound = topk_topp_bound(logits.float(), temperatures.float(),
                                top_ps.float(), top_ks)
        scaled = torch.where(logits.float() >= bound.unsqueeze(1), scaled,
                             torch.full_like(scaled, float("-inf")))
    # Gumbel-max trick: single fused sample, no multinomial sync.
    u = torch.rand(B, V, device=logits.device, generator=generator)
    gumbel = -torch.log(-torch.log(u.clamp_min(1e-20)).clamp_min(1e-20))
    sampled = (scaled + gumbel).argmax(dim=-1)
    greedy_choice = logits.argmax(dim=-1)
    return torch.where(greedy, greedy_choice, sampled)
