"""Checkpoint-loading parity vs HuggingFace transformers (CPU).

Builds tiny random HF checkpoints (Llama and Qwen2 layouts), loads them
through the in-tree safetensors loader, and compares full-precision logits
against the transformers forward. This is the correctness anchor for
real-model serving (the reference delegates this entirely to vLLM —
vllm_worker.py:105-123 — and has no such test; SURVEY §4 calls for
exceeding that)."""

from __future__ import annotations

import json

import pytest
import torch

pytestmark = pytest.mark.integration

transformers = pytest.importorskip("transformers")


def _save_tiny(tmp_path, family="llama"):
    if family == "qwen2":
        cfg = transformers.Qwen2Config(
            vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
            tie_word_embeddings=False,
        )
        model = transformers.Qwen2ForCausalLM(cfg)
    elif family == "mistral":
        cfg = transformers.MistralConfig(
            vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
            sliding_window=32, tie_word_embeddings=False,
        )
        model = transformers.MistralForCausalLM(cfg)
    elif family == "gemma2":
        cfg = transformers.Gemma2Config(
            vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            head_dim=32, max_position_embeddings=256, rope_theta=10000.0,
            rms_norm_eps=1e-6, attn_logit_softcapping=50.0,
            final_logit_softcapping=30.0, sliding_window=64,
            query_pre_attn_scalar=32, tie_word_embeddings=True,
        )
        model = transformers.Gemma2ForCausalLM(cfg)
    else:
        cfg = transformers.LlamaConfig(
            vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
            tie_word_embeddings=False,
        )
        model = transformers.LlamaForCausalLM(cfg)
    model = model.eval().float()
    model.save_pretrained(tmp_path, safe_serialization=True)
    # minimal tokenizer marker so load_tokenizer falls back to bytes
    (tmp_path / "tokenizer_config.json").write_text(json.dumps({}))
    return model


@pytest.mark.parametrize("family", ["llama", "qwen2", "gemma2", "mistral"])
def test_logits_match_transformers(tmp_path, family):
    hf = _save_tiny(tmp_path, family=family)

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine

    engine = LLMEngine(EngineConfig(
        model=str(tmp_path), device="cpu", enforce_eager=True,
        max_num_seqs=2, max_model_len=128, num_kv_blocks=64,
    ))
    spec = engine.spec
    assert spec.num_layers == 2
    assert spec.num_heads == 4 and spec.num_kv_heads == 2
    assert spec.qkv_bias == (family == "qwen2")
    assert spec.post_norms == (family == "gemma2")
    if family == "mistral":
        assert spec.sliding_window == 32
        assert spec.layer_uses_sliding_window(0) and spec.layer_uses_sliding_window(1)

    token_ids = [1, 7, 42, 99, 123, 250, 3]
    # in-tree forward: run a prefill step and capture the logits
    from llmq_amd.engine.scheduler import Sequence
    from llmq_amd.engine.sampling_params import SamplingParams

    seq = Sequence("p", token_ids, SamplingParams(temperature=0.0))
    blocks = engine.allocator.allocate(1)
    seq.block_table = blocks
    runner = engine.runner

    import llmq_amd.engine.model_runner as mr  # noqa: F401

    # reuse execute_prefill but grab logits via compute_logits on last hidden
    dev = engine.device
    ids = torch.tensor(token_ids, dtype=torch.long, device=dev)
    pos = torch.arange(len(token_ids), dtype=torch.long, device=dev)
    from llmq_amd.engine.forward_meta import PrefillMeta

    bs = engine.config.kv_block_size
    slots = [blocks[p // bs] * bs + p % bs for p in range(len(token_ids))]
    meta = PrefillMeta(
        cu_seqlens=torch.tensor([0, len(token_ids)], dtype=torch.int32, device=dev),
        max_seqlen=len(token_ids),
        slot_mapping=torch.tensor(slots, dtype=torch.long, device=dev),
    )
    hidden = engine.model.forward(ids, pos, engine.kv_cache, meta)
    ours = engine.model.compute_logits(hidden)  # [T, vocab]

    with torch.no_grad():
        theirs = hf(torch.tensor([token_ids])).logits[0]

    # fp32 end-to-end on both sides: tight tolerance
    diff = (ours - theirs).abs().max().item()
    ref_scale = theirs.abs().max().item()
    assert diff < 1e-3 * max(1.0, ref_scale), f"logits diverge: {diff}"
