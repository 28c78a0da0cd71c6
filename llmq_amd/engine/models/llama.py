"""Decoder-only transformer covering the Llama / Qwen2 / Gemma-2 families.

One functional implementation driven by ModelSpec switches (SURVEY §2.9:
the model set the reference serves through vLLM). Design notes, MI355X-first:

- Weights are plain tensors (no nn.Module graph): qkv packed into ONE GEMM
  [q+2kv, hidden], gate+up packed into ONE GEMM [2*inter, hidden] — fewer,
  larger hipBLASLt GEMMs (xGMI-budgeted TP shards stay big).
- Hot elementwise/normalisation ops go through llmq_amd.ops → hand-written
  CDNA4 kernels on GPU (fused residual+RMSNorm, fused RoPE q‖k, fused
  SiLU·mul / GeGLU, paged attention, KV scatter).
- Tensor parallelism: column-shard qkv & gate_up, row-shard o & down with a
  single RCCL all-reduce after each of the two row GEMMs per layer (the
  standard 2-allreduce/layer Megatron split; over 7×153 GB/s xGMI links the
  payloads at decode batch sizes are latency-bound, so fewer+larger
  collectives win — SURVEY §2.9 TP note).
"""

from __future__ import annotations

import math
import os
from typing import Dict, List, Optional, Union

import torch
import torch.nn.functional as F

from llmq_amd import ops
from llmq_amd.engine.forward_meta import DecodeMeta, MixedMeta, PrefillMeta
from llmq_amd.engine.kv_cache import KVCache
from llmq_amd.engine.model_specs import ModelSpec
from llmq_amd.parallel import get_tp_group


# Decode projection GEMMs through the hand-written skinny-M kernel
# (ops.skinny_gemm) instead of hipBLASLt. Opt-in while A/B-ing
# (LLMQ_SKINNY_GEMM=1); shapes outside the kernel's envelope fall back.
_USE_SKINNY_GEMM = os.environ.get("LLMQ_SKINNY_GEMM", "0") not in ("0", "false", "")


def _proj(x: torch.Tensor, w: torch.Tensor, bias=None) -> torch.Tensor:
    if (_USE_SKINNY_GEMM and x.is_cuda and x.dtype == torch.bfloat16
            and x.shape[0] <= 1024 and w.shape[1] % 64 == 0
            and w.shape[0] % 4 == 0):
        return ops.skinny_gemm(x, w, bias)
    return F.linear(x, w, bias)


class LayerWeights:
    __slots__ = (
        "input_norm", "qkv", "qkv_bias", "o", "post_attn_norm",
        "pre_mlp_norm", "gate_up", "down", "post_mlp_norm",
    )

    def __init__(self):
        self.input_norm = None
        self.qkv = None
        self.qkv_bias = None
        self.o = None
        self.post_attn_norm = None
        self.pre_mlp_norm = None
        self.gate_up = None
        self.down = None
        self.post_mlp_norm = None


class CausalLM:
    def __init__(
        self,
        spec: ModelSpec,
        device: torch.device,
        dtype: torch.dtype,
        tp_rank: int = 0,
        tp_size: int = 1,
    ):
        self.spec = spec
        self.device = device
        self.dtype = dtype
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        if spec.num_heads % tp_size or spec.num_kv_heads % max(
            1, min(tp_size, spec.num_kv_heads)
        ):
            raise ValueError(
                f"num_heads={spec.num_heads} not divisible by tp_size={tp_size}"
            )
        self.heads = spec.num_heads // tp_size
        # KV heads replicate when tp_size > num_kv_heads is not supported; shard.
        if spec.num_kv_heads % tp_size:
            raise ValueError(
                f"num_kv_heads={spec.num_kv_heads} not divisible by tp_size={tp_size}"
            )
        self.kv_heads = spec.num_kv_heads // tp_size
        self.q_size = self.heads * spec.head_dim
        self.kv_size = self.kv_heads * spec.head_dim
        self.inter = spec.intermediate_size // tp_size
        if spec.intermediate_size % tp_size:
            raise ValueError("intermediate_size not divisible by tp_size")
        # Gemma norms use the (1 + w) convention.
        self.norm_offset = 1.0 if spec.family == "gemma2" else 0.0

        self.embedding: Optional[torch.Tensor] = None
        self.layers: List[LayerWeights] = []
        self.final_norm: Optional[torch.Tensor] = None
        self.lm_head: Optional[torch.Tensor] = None
        self.rope_cache = ops.build_rope_cache(
            spec.max_position_embeddings, spec.head_dim, spec.rope_theta, device,
            rope_scaling=spec.rope_scaling,
        )
        self._alloc()

    # -- weights ---------------------------------------------------------

    def _empty(self, *shape) -> torch.Tensor:
        return torch.empty(*shape, device=self.device, dtype=self.dtype)

    def _alloc(self) -> None:
        s = self.spec
        self.embedding = self._empty(s.vocab_size, s.hidden_size)
        for _ in range(s.num_layers):
            lw = LayerWeights()
            lw.input_norm = self._empty(s.hidden_size)
            lw.qkv = self._empty(self.q_size + 2 * self.kv_size, s.hidden_size)
            if s.qkv_bias:
                lw.qkv_bias = self._empty(self.q_size + 2 * self.kv_size)
            lw.o = self._empty(s.hidden_size, self.q_size)
            lw.pre_mlp_norm = self._empty(s.hidden_size)
            lw.gate_up = self._empty(2 * self.inter, s.hidden_size)
            lw.down = self._empty(s.hidden_size, self.inter)
            if s.post_norms:
                lw.post_attn_norm = self._empty(s.hidden_size)
                lw.post_mlp_norm = self._empty(s.hidden_size)
            self.layers.append(lw)
        self.final_norm = self._empty(s.hidden_size)
        self.lm_head = self.embedding if s.tied_embeddings else self._empty(
            s.vocab_size, s.hidden_size
        )

    @torch.no_grad()
    def random_init(self, seed: int = 0, fast: bool = False) -> None:
        """Random weights for synthetic benchmarking (no network for real
        checkpoints here). Scaled so activations stay finite in bf16.

        fast=True generates on-device (needed for multi-B-param models:
        host-side generation of a 9B model costs ~40 s); fast=False uses a
        CPU generator so CPU and GPU engines with the same seed get
        bit-identical weights (consistency tests)."""
        if fast and self.device.type == "cuda":
            gen = torch.Generator(device=self.device).manual_seed(seed)

            def fill(t: torch.Tensor, std: float) -> None:
                t.normal_(0.0, std, generator=gen)
        else:
            gen = torch.Generator(device="cpu").manual_seed(seed)

            def fill(t: torch.Tensor, std: float) -> None:
                cpu = torch.randn(t.shape, generator=gen, dtype=torch.float32) * std
                t.copy_(cpu.to(t.dtype))

        h = self.spec.hidden_size
        std = 0.02
        out_std = std / math.sqrt(2 * self.spec.num_layers)
        fill(self.embedding, std)
        for lw in self.layers:
            lw.input_norm.fill_(1.0 - self.norm_offset)
            fill(lw.qkv, std)
            if lw.qkv_bias is not None:
                lw.qkv_bias.zero_()
            fill(lw.o, out_std)
            lw.pre_mlp_norm.fill_(1.0 - self.norm_offset)
            fill(lw.gate_up, std)
            fill(lw.down, out_std)
            if lw.post_attn_norm is not None:
                lw.post_attn_norm.fill_(1.0 - self.norm_offset)
                lw.post_mlp_norm.fill_(1.0 - self.norm_offset)
        self.final_norm.fill_(1.0 - self.norm_offset)
        if not self.spec.tied_embeddings:
            fill(self.lm_head, std)

    def named_tensors(self) -> Dict[str, torch.Tensor]:
        out = {"embedding": self.embedding, "final_norm": self.final_norm}
        if not self.spec.tied_embeddings:
            out["lm_head"] = self.lm_head
        for i, lw in enumerate(self.layers):
            p = f"layers.{i}."
            out[p + "input_norm"] = lw.input_norm
            out[p + "qkv"] = lw.qkv
            if lw.qkv_bias is not None:
                out[p + "qkv_bias"] = lw.qkv_bias
            out[p + "o"] = lw.o
            out[p + "pre_mlp_norm"] = lw.pre_mlp_norm
            out[p + "gate_up"] = lw.gate_up
            out[p + "down"] = lw.down
            if lw.post_attn_norm is not None:
                out[p + "post_attn_norm"] = lw.post_attn_norm
                out[p + "post_mlp_norm"] = lw.post_mlp_norm
        return out

    def weight_bytes(self) -> int:
        return sum(t.numel() * t.element_size() for t in self.named_tensors().values())

    # -- forward ---------------------------------------------------------

    def _norm(self, x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
        return ops.rmsnorm(x, w, self.spec.rms_eps, self.norm_offset)

    def _prefill_attn(self, q, k, v, pmeta, kv_cache, layer: int, window: int):
        """Packed varlen attention; chunk continuations assemble their full
        context (gathered past from the paged cache + fresh rows)."""
        s = self.spec
        g = pmeta.gather
        if g is None:
            return ops.varlen_prefill_attention(
                q, k, v, pmeta.cu_seqlens, pmeta.max_seqlen, s.scale,
                s.attn_softcap, window,
            )
        KVH, D = k.shape[1], k.shape[2]
        k_att = q.new_empty(g.total_k, KVH, D)
        v_att = q.new_empty(g.total_k, KVH, D)
        past_k = kv_cache.k[layer][g.blocks, :, g.offs].to(q.dtype)
        past_v = kv_cache.v[layer][g.blocks, :, g.offs].to(q.dtype)
        k_att.index_copy_(0, g.past_dst, past_k)
        v_att.index_copy_(0, g.past_dst, past_v)
        k_att.index_copy_(0, g.fresh_dst, k)
        v_att.index_copy_(0, g.fresh_dst, v)
        return ops.varlen_prefill_attention(
            q, k_att, v_att, pmeta.cu_seqlens, pmeta.max_seqlen, s.scale,
            s.attn_softcap, window, cu_seqlens_k=g.cu_seqlens_k,
        )

    @torch.no_grad()
    def forward(
        self,
        input_ids: torch.Tensor,  # [T]
        positions: torch.Tensor,  # [T]
        kv_cache: KVCache,
        meta: Union[PrefillMeta, DecodeMeta, MixedMeta],
    ) -> torch.Tensor:
        s = self.spec
        tp = get_tp_group() if self.tp_size > 1 else None
        x = F.embedding(input_ids, self.embedding)
        if s.embedding_scale:
            x = x * math.sqrt(s.hidden_size)
            x = x.to(self.dtype)
        residual = x
        h = self._norm(x, self.layers[0].input_norm)
        n_layers = len(self.layers)
        for i, lw in enumerate(self.layers):
            # ---- attention block (h = normed input, set by the previous
            # block's fused norm — launch count is the decode-step bound)
            qkv = _proj(h, lw.qkv, lw.qkv_bias)
            q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)
            T = q.shape[0]
            q = q.view(T, self.heads, s.head_dim)
            k = k.view(T, self.kv_heads, s.head_dim)
            v = v.view(T, self.kv_heads, s.head_dim)
            ops.rope_and_cache(
                q, k, v, kv_cache.k[i], kv_cache.v[i], positions,
                self.rope_cache, meta.slot_mapping,
            )
            window = s.sliding_window if s.layer_uses_sliding_window(i) else 0
            if meta.is_prefill:
                attn = self._prefill_attn(q, k, v, meta, kv_cache, i, window)
            elif meta.is_mixed:
                # per-segment attention; GEMMs/norms already ran packed
                nd = meta.n_decode
                attn_d = ops.paged_decode_attention(
                    q[:nd], kv_cache.k[i], kv_cache.v[i],
                    meta.decode.block_tables, meta.decode.context_lens,
                    s.scale, s.attn_softcap, window,
                )
                attn_p = self._prefill_attn(
                    q[nd:], k[nd:], v[nd:], meta.prefill, kv_cache, i, window
                )
                attn = torch.cat([attn_d, attn_p], dim=0)
            else:
                attn = ops.paged_decode_attention(
                    q, kv_cache.k[i], kv_cache.v[i], meta.block_tables,
                    meta.context_lens, s.scale, s.attn_softcap, window,
                )
            attn_out = _proj(attn.reshape(T, self.q_size), lw.o)
            if tp is not None:
                attn_out = tp.all_reduce(attn_out)
            if s.post_norms:
                # Gemma-2 sandwich fused: residual += norm(attn_out, post);
                # h = norm(residual, pre_mlp)
                h, residual = ops.norm_add_norm(
                    attn_out, residual, lw.post_attn_norm, lw.pre_mlp_norm,
                    s.rms_eps, self.norm_offset,
                )
            else:
                h, residual = ops.fused_add_rmsnorm(
                    attn_out, residual, lw.pre_mlp_norm, s.rms_eps, self.norm_offset
                )
            # ---- MLP block
            gate_up = _proj(h, lw.gate_up)
            act = ops.gelu_tanh_and_mul(gate_up) if s.gelu else ops.silu_and_mul(gate_up)
            mlp_out = _proj(act, lw.down)
            if tp is not None:
                mlp_out = tp.all_reduce(mlp_out)
            # The next block's input norm (or the final norm) fuses with this
            # block's residual add (+ post norm for Gemma-2).
            w_next = (
                self.layers[i + 1].input_norm if i + 1 < n_layers else self.final_norm
            )
            if s.post_norms:
                h, residual = ops.norm_add_norm(
                    mlp_out, residual, lw.post_mlp_norm, w_next,
                    s.rms_eps, self.norm_offset,
                )
            else:
                h, residual = ops.fused_add_rmsnorm(
                    mlp_out, residual, w_next, s.rms_eps, self.norm_offset
                )
        return h

    @torch.no_grad()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        """hidden [N, hidden] → logits [N, vocab] (fp32)."""
        logits = _proj(hidden, self.lm_head).float()
        cap = self.spec.final_softcap
        if cap and cap > 0:
            logits = torch.tanh(logits / cap) * cap
        return logits
