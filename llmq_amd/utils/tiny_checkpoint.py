"""Build a tiny REAL-layout checkpoint offline (testing utility).

Produces what a HuggingFace hub download would: config.json +
model.safetensors (Llama/Qwen2/Gemma-2 layout) + a real ``tokenizer.json``
(WordLevel, built with the `tokenizers` library) + tokenizer_config.json
with a chat template. Lets GPU boxes (no network) exercise the FULL
serving path the reference gets from real checkpoints: load from disk →
HF tokenize → chat-template → generate → non-empty detokenized text
(reference consumes this via vllm_worker.py:105-123,146,175-177).
"""

from __future__ import annotations

import json
from pathlib import Path


def build_tiny_checkpoint(
    path,
    family: str = "llama",
    vocab_size: int = 512,
    hidden: int = 512,
    layers: int = 2,
    heads: int = 8,
    kv_heads: int = 4,
    seed: int = 0,
) -> str:
    """Write a tiny random-weight checkpoint + real tokenizer; returns path.

    head_dim = hidden/heads must be a GPU-kernel-supported shape
    (64/128/256) — the default 512/8 gives 64."""
    import torch
    import transformers
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    path = Path(path)
    path.mkdir(parents=True, exist_ok=True)
    torch.manual_seed(seed)

    if family == "llama":
        cfg = transformers.LlamaConfig(
            vocab_size=vocab_size, hidden_size=hidden, intermediate_size=hidden * 2,
            num_hidden_layers=layers, num_attention_heads=heads,
            num_key_value_heads=kv_heads, max_position_embeddings=2048,
            rope_theta=10000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
            bos_token_id=1, eos_token_id=2,
        )
        model = transformers.LlamaForCausalLM(cfg)
    elif family == "qwen2":
        cfg = transformers.Qwen2Config(
            vocab_size=vocab_size, hidden_size=hidden, intermediate_size=hidden * 2,
            num_hidden_layers=layers, num_attention_heads=heads,
            num_key_value_heads=kv_heads, max_position_embeddings=2048,
            rope_theta=10000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
            bos_token_id=1, eos_token_id=2,
        )
        model = transformers.Qwen2ForCausalLM(cfg)
    elif family == "gemma2":
        cfg = transformers.Gemma2Config(
            vocab_size=vocab_size, hidden_size=hidden, intermediate_size=hidden * 2,
            num_hidden_layers=layers, num_attention_heads=heads,
            num_key_value_heads=kv_heads, head_dim=hidden // heads,
            max_position_embeddings=2048, rms_norm_eps=1e-6,
            bos_token_id=1, eos_token_id=2,
        )
        model = transformers.Gemma2ForCausalLM(cfg)
    else:
        raise ValueError(f"unknown family {family}")
    model.eval().float().save_pretrained(path, safe_serialization=True)

    # Real tokenizer.json: WordLevel over a synthetic word vocabulary.
    vocab = {"<unk>": 0, "<s>": 1, "</s>": 2}
    words = ["hello", "world", "translate", "the", "quick", "brown", "fox",
             "jumps", "over", "lazy", "dog", "model", "token", "queue"]
    for w in words:
        if len(vocab) < vocab_size:
            vocab.setdefault(w, len(vocab))
    i = 0
    while len(vocab) < vocab_size:
        vocab[f"w{i}"] = len(vocab)
        i += 1
    tok = Tokenizer(WordLevel(vocab, unk_token="<unk>"))
    tok.pre_tokenizer = Whitespace()
    tok.save(str(path / "tokenizer.json"))
    (path / "tokenizer_config.json").write_text(json.dumps({
        "tokenizer_class": "PreTrainedTokenizerFast",
        "bos_token": "<s>",
        "eos_token": "</s>",
        "unk_token": "<unk>",
        "model_max_length": 2048,
        "chat_template": (
            "{% for message in messages %}"
            "{{ message['role'] }} : {{ message['content'] }}\n"
            "{% endfor %}"
            "{% if add_generation_prompt %}assistant :{% endif %}"
        ),
    }, indent=1))
    (path / "special_tokens_map.json").write_text(json.dumps({
        "bos_token": "<s>", "eos_token": "</s>", "unk_token": "<unk>",
    }))
    return str(path)
