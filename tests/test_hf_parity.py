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


def test_llama3_rope_scaling_matches_transformers(tmp_path):
    """Llama-3.1/3.2 checkpoints ship rope_scaling (rope_type 'llama3');
    logits must match transformers' wavelength-remapped RoPE, and a config
    WITHOUT scaling must differ from one with it (the test has teeth)."""
    cfg = transformers.LlamaConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
        tie_word_embeddings=False,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 32},
    )
    torch.manual_seed(5)
    hf = transformers.LlamaForCausalLM(cfg).eval().float()
    hf.save_pretrained(tmp_path, safe_serialization=True)
    (tmp_path / "tokenizer_config.json").write_text(json.dumps({}))

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine

    eng = LLMEngine(EngineConfig(
        model=str(tmp_path), device="cpu", enforce_eager=True,
        num_kv_blocks=64, max_model_len=256,
    ))
    assert eng.spec.rope_scaling["rope_type"] == "llama3"
    ids = torch.randint(0, 300, (48,)).tolist()
    with torch.no_grad():
        ref = hf(torch.tensor([ids])).logits[0]
    ours = eng.model.forward_logits_all(torch.tensor(ids)) \
        if hasattr(eng.model, "forward_logits_all") else None
    if ours is None:
        # drive through the engine: greedy continuation must match HF's
        from llmq_amd.engine.sampling_params import SamplingParams
        eng.add_request("r", prompt_token_ids=ids,
                        params=SamplingParams(temperature=0.0, max_tokens=6,
                                              ignore_eos=True))
        toks = []
        while eng.has_unfinished():
            for out in eng.step():
                toks.extend(out.new_token_ids)
        hf_toks = []
        cur = list(ids)
        for _ in range(6):
            with torch.no_grad():
                nxt = int(hf(torch.tensor([cur])).logits[0, -1].argmax())
            hf_toks.append(nxt)
            cur.append(nxt)
        assert toks == hf_toks, (toks, hf_toks)

    # teeth: our inv_freq remap must equal transformers' llama3 rope init
    # exactly (greedy chains can coincide at tiny geometry; tables cannot)
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

    from llmq_amd.ops.torch_ref import build_rope_cache

    inv_ref, scale = ROPE_INIT_FUNCTIONS["llama3"](cfg, device="cpu")
    assert scale == 1.0
    sc = dict(getattr(cfg, "rope_scaling", None) or cfg.rope_parameters)
    tab = build_rope_cache(256, 16, 10000.0, "cpu", rope_scaling=sc)
    t = torch.arange(256, dtype=torch.float32)
    ref = torch.cat([torch.outer(t, inv_ref).cos(),
                     torch.outer(t, inv_ref).sin()], -1)
    assert torch.equal(tab, ref), (tab - ref).abs().max()
