"""Model architecture specs for the Llama/Qwen2/Gemma-2 families.

The reference serves Tower-Plus-{2B,9B,72B}, Llama-3.2 and Gemma-2 via vLLM
(README.md:98-108,169). All of those are decoder-only
RMSNorm+RoPE+GQA transformers differing in a handful of switches, captured
here as one ModelSpec:

  qkv_bias          Qwen2 adds bias to q/k/v projections
  tied_embeddings   small models tie lm_head to the embedding
  gelu              Gemma uses GeGLU (gelu_tanh) instead of SiLU
  post_norms        Gemma-2 adds post-attention/post-mlp norms
  attn_softcap /    Gemma-2 logit soft-capping
  final_softcap
  sliding_window    Gemma-2 alternates local/global attention layers
  embedding_scale   Gemma scales embeddings by sqrt(hidden)

A spec can be loaded from a HF checkpoint directory's config.json or picked
from the built-in presets by name (synthetic/random-init benchmarking —
no network in this environment).
"""

from __future__ import annotations

import json
import math
from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional


@dataclass
class ModelSpec:
    name: str
    family: str  # "llama" | "qwen2" | "gemma2"
    vocab_size: int
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    rms_eps: float = 1e-6
    rope_theta: float = 10000.0
    max_position_embeddings: int = 8192
    qkv_bias: bool = False
    tied_embeddings: bool = False
    gelu: bool = False
    post_norms: bool = False
    attn_softcap: float = 0.0
    final_softcap: float = 0.0
    sliding_window: int = 0  # 0 = none; Gemma-2: applied on even layers
    embedding_scale: bool = False
    attn_scale: Optional[float] = None  # default 1/sqrt(head_dim)
    rope_scaling: Optional[dict] = None  # HF rope_scaling (llama3/linear)
    eos_token_id: int = 2
    bos_token_id: int = 1

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    @property
    def scale(self) -> float:
        if self.attn_scale is not None:
            return self.attn_scale
        return 1.0 / math.sqrt(self.head_dim)

    def layer_uses_sliding_window(self, layer_idx: int) -> bool:
        if self.sliding_window <= 0:
            return False
        # Gemma-2 alternates local/global layers; Mistral windows every layer.
        if self.family == "gemma2":
            return layer_idx % 2 == 0
        return True

    def param_count(self) -> int:
        embed = self.vocab_size * self.hidden_size
        per_layer = (
            self.hidden_size * (self.q_size + 2 * self.kv_size)  # qkv
            + self.q_size * self.hidden_size  # o
            + 3 * self.hidden_size * self.intermediate_size  # gate/up/down
            + 2 * self.hidden_size  # norms
        )
        if self.post_norms:
            per_layer += 2 * self.hidden_size
        total = embed + self.num_layers * per_layer + self.hidden_size
        if not self.tied_embeddings:
            total += embed
        return total


def _llama(name, vocab, hidden, inter, layers, heads, kv_heads, head_dim=None, **kw):
    return ModelSpec(
        name=name, family="llama", vocab_size=vocab, hidden_size=hidden,
        intermediate_size=inter, num_layers=layers, num_heads=heads,
        num_kv_heads=kv_heads, head_dim=head_dim or hidden // heads,
        rms_eps=1e-5, **kw,
    )


def _qwen2(name, vocab, hidden, inter, layers, heads, kv_heads, head_dim=None, **kw):
    return ModelSpec(
        name=name, family="qwen2", vocab_size=vocab, hidden_size=hidden,
        intermediate_size=inter, num_layers=layers, num_heads=heads,
        num_kv_heads=kv_heads, head_dim=head_dim or hidden // heads, rms_eps=1e-6,
        rope_theta=1000000.0, qkv_bias=True, **kw,
    )


def _gemma2(name, vocab, hidden, inter, layers, heads, kv_heads, head_dim, **kw):
    kw.setdefault("sliding_window", 4096)
    return ModelSpec(
        name=name, family="gemma2", vocab_size=vocab, hidden_size=hidden,
        intermediate_size=inter, num_layers=layers, num_heads=heads,
        num_kv_heads=kv_heads, head_dim=head_dim, rms_eps=1e-6,
        gelu=True, post_norms=True, attn_softcap=50.0, final_softcap=30.0,
        embedding_scale=True, tied_embeddings=True,
        eos_token_id=1, bos_token_id=2, **kw,
    )


# Llama-3.1/3.2 ship rope_scaling in config.json (rope_type "llama3");
# real checkpoints are wrong without it, so the presets carry it too.
_LLAMA31_ROPE = {"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
                 "high_freq_factor": 4.0,
                 "original_max_position_embeddings": 8192}
_LLAMA32_ROPE = {"rope_type": "llama3", "factor": 32.0, "low_freq_factor": 1.0,
                 "high_freq_factor": 4.0,
                 "original_max_position_embeddings": 8192}

PRESETS = {
    # tiny configs for CPU tests
    # head_dim 64 (not hidden/heads = 16): the HIP attention kernels support
    # D in {64, 128, 256}, and the GPU consistency tests run these presets.
    "tiny-llama": _llama("tiny-llama", 512, 64, 128, 2, 4, 2, head_dim=64,
                         rope_theta=10000.0,
                         max_position_embeddings=512, tied_embeddings=True),
    "tiny-llama-d128": _llama("tiny-llama-d128", 512, 64, 128, 2, 4, 2,
                              head_dim=128, rope_theta=10000.0,
                              max_position_embeddings=512, tied_embeddings=True),
    "tiny-qwen2": _qwen2("tiny-qwen2", 512, 64, 128, 2, 4, 2, head_dim=64,
                         max_position_embeddings=512, tied_embeddings=True),
    "tiny-gemma2": _gemma2("tiny-gemma2", 512, 64, 128, 2, 4, 2, 64,
                           sliding_window=64, max_position_embeddings=512),
    # Llama 3.2 (vocab 128256, rope theta 500k)
    "llama-3.2-1b": _llama("llama-3.2-1b", 128256, 2048, 8192, 16, 32, 8,
                           head_dim=64, rope_theta=500000.0,
                           rope_scaling=_LLAMA32_ROPE,
                           max_position_embeddings=131072, tied_embeddings=True,
                           eos_token_id=128001, bos_token_id=128000),
    "llama-3.2-3b": _llama("llama-3.2-3b", 128256, 3072, 8192, 28, 24, 8,
                           head_dim=128, rope_theta=500000.0,
                           rope_scaling=_LLAMA32_ROPE,
                           max_position_embeddings=131072, tied_embeddings=True,
                           eos_token_id=128001, bos_token_id=128000),
    # Qwen2.5 family (Tower-Plus-2B = Qwen2.5-1.5B arch; -72B = Qwen2.5-72B)
    "qwen2.5-1.5b": _qwen2("qwen2.5-1.5b", 151936, 1536, 8960, 28, 12, 2,
                           max_position_embeddings=32768, tied_embeddings=True,
                           eos_token_id=151645, bos_token_id=151643),
    "qwen2.5-7b": _qwen2("qwen2.5-7b", 152064, 3584, 18944, 28, 28, 4,
                         max_position_embeddings=32768,
                         eos_token_id=151645, bos_token_id=151643),
    "qwen2.5-72b": _qwen2("qwen2.5-72b", 152064, 8192, 29568, 80, 64, 8,
                          max_position_embeddings=32768,
                          eos_token_id=151645, bos_token_id=151643),
    "qwen2.5-32b": _qwen2("qwen2.5-32b", 152064, 5120, 27648, 64, 40, 8,
                          max_position_embeddings=32768,
                          eos_token_id=151645, bos_token_id=151643),
    # Llama-3.1-8B / 70B (same layout; 70B pairs with TP=8 or fits 1 GPU)
    "llama-3.1-8b": _llama("llama-3.1-8b", 128256, 4096, 14336, 32, 32, 8,
                           head_dim=128, rope_theta=500000.0,
                           rope_scaling=_LLAMA31_ROPE,
                           max_position_embeddings=131072,
                           eos_token_id=128001, bos_token_id=128000),
    "llama-3.1-70b": _llama("llama-3.1-70b", 128256, 8192, 28672, 80, 64, 8,
                            head_dim=128, rope_theta=500000.0,
                            rope_scaling=_LLAMA31_ROPE,
                            max_position_embeddings=131072,
                            eos_token_id=128001, bos_token_id=128000),
    # Mistral-7B (llama layout, 4096-token sliding window on every layer)
    "mistral-7b": _llama("mistral-7b", 32000, 4096, 14336, 32, 32, 8,
                         rope_theta=10000.0, max_position_embeddings=32768,
                         sliding_window=4096, eos_token_id=2, bos_token_id=1),
    # Gemma-2 (Tower-Plus-9B is built on Gemma-2-9B)
    "gemma-2-9b": _gemma2("gemma-2-9b", 256000, 3584, 14336, 42, 16, 8, 256,
                          max_position_embeddings=8192),
    # 27b: query_pre_attn_scalar = hidden/heads = 144 (NOT head_dim as on
    # 9b) -> attention scale 1/sqrt(144)
    "gemma-2-27b": _gemma2("gemma-2-27b", 256000, 4608, 36864, 46, 32, 16, 128,
                           attn_scale=144.0 ** -0.5,
                           max_position_embeddings=8192),
}

# Aliases for the model names the reference's production configs use.
ALIASES = {
    "tower-plus-2b": "qwen2.5-1.5b",
    "unbabel/tower-plus-2b": "qwen2.5-1.5b",
    "tower-plus-9b": "gemma-2-9b",
    "unbabel/tower-plus-9b": "gemma-2-9b",
    "tower-plus-72b": "qwen2.5-72b",
    "unbabel/tower-plus-72b": "qwen2.5-72b",
    "llama-3.2-1b-instruct": "llama-3.2-1b",
    "meta-llama/llama-3.2-1b-instruct": "llama-3.2-1b",
    "llama-3.2-3b-instruct": "llama-3.2-3b",
    "meta-llama/llama-3.2-3b-instruct": "llama-3.2-3b",
    "google/gemma-2-9b": "gemma-2-9b",
    "google/gemma-2-9b-it": "gemma-2-9b",
    "google/gemma-2-27b": "gemma-2-27b",
    "meta-llama/llama-3.1-8b-instruct": "llama-3.1-8b",
    "llama-3.1-8b-instruct": "llama-3.1-8b",
    "meta-llama/llama-3.1-70b-instruct": "llama-3.1-70b",
    "llama-3.1-70b-instruct": "llama-3.1-70b",
    "qwen/qwen2.5-32b-instruct": "qwen2.5-32b",
}


def spec_from_hf_config(path: Path, name: str) -> ModelSpec:
    """Build a spec from a HF checkpoint directory's config.json."""
    with open(path / "config.json", "r", encoding="utf-8") as fh:
        cfg = json.load(fh)
    arch = (cfg.get("architectures") or [""])[0].lower()
    model_type = cfg.get("model_type", "").lower()
    if "gemma2" in arch or model_type == "gemma2":
        family = "gemma2"
    elif "qwen2" in arch or model_type == "qwen2":
        family = "qwen2"
    elif "mistral" in arch or model_type == "mistral":
        family = "mistral"  # llama layout + sliding window on every layer
    else:
        family = "llama"
    heads = cfg["num_attention_heads"]
    hidden = cfg["hidden_size"]
    head_dim = cfg.get("head_dim") or hidden // heads
    # rope config: transformers <=4.x puts rope_theta at top level and the
    # scaling dict under "rope_scaling"; transformers 5.x nests BOTH under
    # "rope_parameters". Normalise to (theta, scaling-or-None).
    rope_params = cfg.get("rope_scaling") or cfg.get("rope_parameters") or {}
    rope_theta = cfg.get("rope_theta") or rope_params.get("rope_theta") or 10000.0
    rope_type = rope_params.get("rope_type") or rope_params.get("type")
    rope_scaling = dict(rope_params) if rope_type not in (None, "default") else None
    spec = ModelSpec(
        name=name,
        family=family,
        vocab_size=cfg["vocab_size"],
        hidden_size=hidden,
        intermediate_size=cfg["intermediate_size"],
        num_layers=cfg["num_hidden_layers"],
        num_heads=heads,
        num_kv_heads=cfg.get("num_key_value_heads", heads),
        head_dim=head_dim,
        rms_eps=cfg.get("rms_norm_eps", 1e-6),
        rope_theta=rope_theta,
        rope_scaling=rope_scaling,
        max_position_embeddings=cfg.get("max_position_embeddings", 8192),
        qkv_bias=family == "qwen2",
        tied_embeddings=cfg.get("tie_word_embeddings", False),
        gelu=family == "gemma2",
        post_norms=family == "gemma2",
        attn_softcap=cfg.get("attn_logit_softcapping") or 0.0,
        final_softcap=cfg.get("final_logit_softcapping") or 0.0,
        sliding_window=(
            (cfg.get("sliding_window") or 0) if family in ("gemma2", "mistral") else 0
        ),
        embedding_scale=family == "gemma2",
        attn_scale=(
            cfg.get("query_pre_attn_scalar") and 1.0 / math.sqrt(cfg["query_pre_attn_scalar"])
        ),
        eos_token_id=_first(cfg.get("eos_token_id"), 2),
        bos_token_id=_first(cfg.get("bos_token_id"), 1),
    )
    return spec


def _first(x, default: int = 2) -> int:
    # HF configs may carry an int, a list, or null (e.g. Qwen2 bos_token_id).
    if x is None:
        return default
    if isinstance(x, list):
        return int(x[0]) if x else default
    return int(x)


def resolve_spec(model: str) -> ModelSpec:
    """Resolve a model name or checkpoint path to a spec."""
    path = Path(model)
    if path.is_dir() and (path / "config.json").is_file():
        return spec_from_hf_config(path, path.name)
    key = model.lower()
    key = ALIASES.get(key, key)
    if key in PRESETS:
        return PRESETS[key]
    raise ValueError(
        f"Unknown model '{model}'. Provide a checkpoint directory with config.json "
        f"or one of: {sorted(PRESETS) + sorted(ALIASES)}"
    )
