"""Reference implementations of the engine's hot ops in plain PyTorch.

These are the NUMERICS ORACLES for the HIP kernels (tests compare the
CDNA4 kernels against these at fp32) and the CPU execution path for tests.
They are deliberately simple; nothing here runs in the GPU hot loop.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float, offset: float = 0.0) -> torch.Tensor:
    """RMSNorm. Gemma uses weight offset +1 (x * (1 + w))."""
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    xf = xf * torch.rsqrt(var + eps)
    return (xf * (weight.float() + offset)).to(dtype)


def norm_add_norm(x, residual, w_post, w_pre, eps, offset=0.0):
    """residual += rmsnorm(x, w_post); x = rmsnorm(residual, w_pre)."""
    t = rmsnorm(x, w_post, eps, offset)
    residual = (residual.float() + t.float()).to(x.dtype)
    out = rmsnorm(residual, w_pre, eps, offset)
    return out, residual


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float, offset: float = 0.0
) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual += x; out = rmsnorm(residual). Returns (out, new_residual)."""
    new_residual = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps, offset), new_residual


def build_rope_cache(
    max_positions: int, rotary_dim: int, theta: float, device,
    dtype=torch.float32, rope_scaling=None,
) -> torch.Tensor:
    """[max_positions, rotary_dim]: first half cos, second half sin.

    rope_scaling: HF config.json dict. Supported: "llama3" (Llama-3.1/3.2
    wavelength remap, transformers modeling_rope_utils
    _compute_llama3_parameters) and "linear" (position interpolation —
    dividing inv_freq by factor is identical to dividing positions).
    Unknown types raise rather than silently serving wrong rotations.
    """
    import math as _math

    inv_freq = 1.0 / (
        theta ** (torch.arange(0, rotary_dim, 2, device=device, dtype=torch.float32) / rotary_dim)
    )
    if rope_scaling:
        rtype = rope_scaling.get("rope_type") or rope_scaling.get("type")
        factor = float(rope_scaling.get("factor", 1.0))
        if rtype == "llama3":
            low = float(rope_scaling["low_freq_factor"])
            high = float(rope_scaling["high_freq_factor"])
            orig = float(rope_scaling["original_max_position_embeddings"])
            wavelen = 2 * _math.pi / inv_freq
            smooth = (orig / wavelen - low) / (high - low)
            smoothed = (1 - smooth) * inv_freq / factor + smooth * inv_freq
            inv_freq = torch.where(
                wavelen < orig / high, inv_freq,
                torch.where(wavelen > orig / low, inv_freq / factor, smoothed),
            )
        elif rtype == "linear":
            inv_freq = inv_freq / factor
        elif rtype in (None, "default"):
            pass
        else:
            raise ValueError(f"unsupported rope_scaling type: {rtype!r}")
    t = torch.arange(max_positions, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [P, rotary_dim/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)


def rope_inplace(
    q: torch.Tensor,  # [T, num_heads, head_dim]
    k: torch.Tensor,  # [T, num_kv_heads, head_dim]
    positions: torch.Tensor,  # [T]
    cos_sin: torch.Tensor,  # [max_pos, head_dim] (cos || sin)
) -> None:
    """NeoX-style (rotate-half) RoPE applied in place to q and k."""
    half = q.shape[-1] // 2
    cs = cos_sin[positions]  # [T, head_dim]
    cos = cs[:, :half].unsqueeze(1)  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)
    for t in (q, k):
        # clone: for f32 inputs .float() aliases, and the first write below
        # would corrupt x1 before the second line reads it
        x1 = t[..., :half].float().clone()
        x2 = t[..., half:].float().clone()
        t[..., :half] = (x1 * cos - x2 * sin).to(t.dtype)
        t[..., half:] = (x2 * cos + x1 * sin).to(t.dtype)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x = [*, 2*d]: silu(x[..., :d]) * x[..., d:]."""
    d = x.shape[-1] // 2
    a, b = x[..., :d], x[..., d:]
    return (torch.nn.functional.silu(a.float()) * b.float()).to(x.dtype)


def gelu_tanh_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    a, b = x[..., :d], x[..., d:]
    return (torch.nn.functional.gelu(a.float(), approximate="tanh") * b.float()).to(x.dtype)


def reshape_and_cache(
    key: torch.Tensor,  # [T, kv_heads, head_dim]
    value: torch.Tensor,
    k_cache: torch.Tensor,  # [num_blocks, kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [T] int32/int64 global slot ids
) -> None:
    block_size = k_cache.shape[2]
    slots = slot_mapping.long()
    blocks = slots // block_size
    offs = slots % block_size
    k_cache[blocks, :, offs, :] = key.to(k_cache.dtype)
    v_cache[blocks, :, offs, :] = value.to(v_cache.dtype)


def _softcap(scores: torch.Tensor, cap: float) -> torch.Tensor:
    if cap and cap > 0:
        return torch.tanh(scores / cap) * cap
    return scores


def paged_decode_attention(
    q: torch.Tensor,  # [B, num_heads, head_dim]
    k_cache: torch.Tensor,  # [num_blocks, kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks] int32
    context_lens: torch.Tensor,  # [B] int32 (length INCLUDING current token)
    scale: float,
    softcap: float = 0.0,
    window: int = 0,
) -> torch.Tensor:
    """Single-token decode attention over the paged cache (reference)."""
    B, H, D = q.shape
    KVH = k_cache.shape[1]
    bs = k_cache.shape[2]
    group = H // KVH
    out = torch.empty_like(q)
    for b in range(B):
        L = int(context_lens[b])
        nblocks = (L + bs - 1) // bs
        blocks = block_tables[b, :nblocks].long()
        k = k_cache[blocks].permute(1, 0, 2, 3).reshape(KVH, nblocks * bs, D)[:, :L]
        v = v_cache[blocks].permute(1, 0, 2, 3).reshape(KVH, nblocks * bs, D)[:, :L]
        start = max(0, L - window) if window else 0
        k, v = k[:, start:], v[:, start:]
        qb = q[b].view(KVH, group, D).float()  # [KVH, G, D]
        scores = torch.einsum("hgd,hld->hgl", qb, k.float()) * scale
        scores = _softcap(scores, softcap)
        probs = torch.softmax(scores, dim=-1)
        ob = torch.einsum("hgl,hld->hgd", probs, v.float())
        out[b] = ob.reshape(H, D).to(q.dtype)
    return out


def varlen_prefill_attention(
    q: torch.Tensor,  # [Tq, num_heads, head_dim]
    k: torch.Tensor,  # [Tk, kv_heads, head_dim]
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,  # [B+1] query offsets
    scale: float,
    softcap: float = 0.0,
    window: int = 0,
    cu_seqlens_k: "Optional[torch.Tensor]" = None,  # [B+1] key offsets
) -> torch.Tensor:
    """Causal attention over packed variable-length sequences (reference).

    With cu_seqlens_k, each sequence's K/V may be LONGER than its Q (chunked
    prefill: the queries are the last Tq positions of a Tk-long context);
    query row i has absolute position (Tk - Tq) + i."""
    T, H, D = q.shape
    KVH = k.shape[1]
    group = H // KVH
    cu_k = cu_seqlens if cu_seqlens_k is None else cu_seqlens_k
    out = torch.empty_like(q)
    for b in range(len(cu_seqlens) - 1):
        s, e = int(cu_seqlens[b]), int(cu_seqlens[b + 1])
        sk, ek = int(cu_k[b]), int(cu_k[b + 1])
        Lq, Lk = e - s, ek - sk
        off = Lk - Lq  # absolute position of q row 0
        qb = q[s:e].float()  # [Lq, H, D]
        kb = k[sk:ek].float().repeat_interleave(group, dim=1)  # [Lk, H, D]
        vb = v[sk:ek].float().repeat_interleave(group, dim=1)
        scores = torch.einsum("ihd,jhd->hij", qb, kb) * scale
        scores = _softcap(scores, softcap)
        i = torch.arange(Lq, device=q.device).view(-1, 1) + off
        j = torch.arange(Lk, device=q.device).view(1, -1)
        mask = j > i
        if window:
            mask = mask | (j <= i - window)
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        ob = torch.einsum("hij,jhd->ihd", probs, vb)
        out[s:e] = ob.to(q.dtype)
    return out


def sample_tokens(
    logits: torch.Tensor,  # [B, vocab] float
    temperatures: torch.Tensor,  # [B]
    top_ps: torch.Tensor,  # [B]
    top_ks: torch.Tensor,  # [B] int (0 = off)
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Temperature / top-k / top-p sampling; greedy where temperature == 0."""
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    greedy_mask = temperatures <= 0
    if greedy_mask.any():
        out[greedy_mask] = logits[greedy_mask].argmax(dim=-1)
    sample_idx = (~greedy_mask).nonzero(as_tuple=True)[0]
    for i in sample_idx.tolist():
        l = logits[i].float() / float(temperatures[i])
        k = int(top_ks[i])
        if k > 0 and k < V:
            kth = torch.topk(l, k).values[-1]
            l = torch.where(l < kth, torch.full_like(l, float("-inf")), l)
        p = float(top_ps[i])
        if p < 1.0:
            sorted_l, sorted_i = torch.sort(l, descending=True)
            probs = torch.softmax(sorted_l, dim=-1)
            cum = torch.cumsum(probs, dim=-1)
            cut = (cum - probs) >= p  # keep tokens whose preceding mass < p
            sorted_l[cut] = float("-inf")
            l = torch.full_like(l, float("-inf")).scatter(0, sorted_i, sorted_l)
        probs = torch.softmax(l, dim=-1)
        out[i] = torch.multinomial(probs, 1, generator=generator).squeeze(0)
    return out
