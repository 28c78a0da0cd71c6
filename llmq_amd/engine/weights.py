"""Safetensors checkpoint loading (HF layout → packed in-tree layout).

Reference counterpart: vLLM's weight loader, consumed implicitly through
AsyncLLMEngine.from_engine_args (vllm_worker.py:105-123). Handles the
Llama/Qwen2/Gemma-2 HF naming and packs q/k/v and gate/up into the fused
GEMM tensors. Tensor-parallel loads take each rank's shard only.
"""

from __future__ import annotations

import logging
from pathlib import Path
from typing import Dict, List

import torch

logger = logging.getLogger(__name__)


def _shard_rows(t: torch.Tensor, rank: int, size: int) -> torch.Tensor:
    if size == 1:
        return t
    n = t.shape[0] // size
    return t[rank * n : (rank + 1) * n]


def _shard_cols(t: torch.Tensor, rank: int, size: int) -> torch.Tensor:
    if size == 1:
        return t
    n = t.shape[1] // size
    return t[:, rank * n : (rank + 1) * n]


def load_safetensors_weights(model, path: Path) -> None:
    from safetensors import safe_open  # noqa: PLC0415

    spec = model.spec
    rank, size = model.tp_rank, model.tp_size
    files = sorted(path.glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")

    # name → file handle lazily
    tensors: Dict[str, torch.Tensor] = {}
    for f in files:
        with safe_open(str(f), framework="pt", device="cpu") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    def get(name: str) -> torch.Tensor:
        for candidate in (name, "model." + name):
            if candidate in tensors:
                return tensors[candidate]
        raise KeyError(f"missing tensor {name} (have {len(tensors)})")

    def copy_(dst: torch.Tensor, src: torch.Tensor) -> None:
        if dst.shape != src.shape:
            raise ValueError(f"shape mismatch: {tuple(dst.shape)} vs {tuple(src.shape)}")
        dst.copy_(src.to(dst.dtype))

    copy_(model.embedding, get("embed_tokens.weight"))
    copy_(model.final_norm, get("norm.weight"))
    if not spec.tied_embeddings:
        copy_(model.lm_head, tensors.get("lm_head.weight", get("embed_tokens.weight")))

    for i, lw in enumerate(model.layers):
        p = f"layers.{i}."
        copy_(lw.input_norm, get(p + "input_layernorm.weight"))
        q = _shard_rows(get(p + "self_attn.q_proj.weight"), rank, size)
        k = _shard_rows(get(p + "self_attn.k_proj.weight"), rank, size)
        v = _shard_rows(get(p + "self_attn.v_proj.weight"), rank, size)
        copy_(lw.qkv, torch.cat([q, k, v], dim=0))
        if lw.qkv_bias is not None:
            qb = _shard_rows(get(p + "self_attn.q_proj.bias"), rank, size)
            kb = _shard_rows(get(p + "self_attn.k_proj.bias"), rank, size)
            vb = _shard_rows(get(p + "self_attn.v_proj.bias"), rank, size)
            copy_(lw.qkv_bias, torch.cat([qb, kb, vb], dim=0))
        copy_(lw.o, _shard_cols(get(p + "self_attn.o_proj.weight"), rank, size))
        gate = _shard_rows(get(p + "mlp.gate_proj.weight"), rank, size)
        up = _shard_rows(get(p + "mlp.up_proj.weight"), rank, size)
        copy_(lw.gate_up, torch.cat([gate, up], dim=0))
        copy_(lw.down, _shard_cols(get(p + "mlp.down_proj.weight"), rank, size))
        if spec.post_norms:
            copy_(lw.post_attn_norm, get(p + "post_attention_layernorm.weight"))
            copy_(lw.pre_mlp_norm, get(p + "pre_feedforward_layernorm.weight"))
            copy_(lw.post_mlp_norm, get(p + "post_feedforward_layernorm.weight"))
        else:
            copy_(lw.pre_mlp_norm, get(p + "post_attention_layernorm.weight"))
    logger.info("loaded %d tensors from %s", len(tensors), path)
