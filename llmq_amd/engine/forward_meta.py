"""Per-step attention metadata passed to the model forward."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class PrefillMeta:
    """Packed variable-length prompt batch."""

    cu_seqlens: torch.Tensor  # [B+1] int32
    max_seqlen: int
    slot_mapping: torch.Tensor  # [T] int64 global KV slots

    is_prefill: bool = True
    is_mixed: bool = False


@dataclass
class DecodeMeta:
    """Single-token decode batch over the paged cache."""

    block_tables: torch.Tensor  # [B, max_blocks] int32
    context_lens: torch.Tensor  # [B] int32 (including the new token)
    slot_mapping: torch.Tensor  # [B] int64

    is_prefill: bool = False
    is_mixed: bool = False


@dataclass
class MixedMeta:
    """One step carrying BOTH a decode batch (rows [0, n_decode)) and a
    packed prefill batch (rows [n_decode, T)) — decode never stalls behind
    admissions. Attention runs per segment; everything else (norms, GEMMs,
    rope+cache) runs on the packed whole."""

    n_decode: int
    decode: "DecodeMeta"
    prefill: "PrefillMeta"
    slot_mapping: torch.Tensor  # [T] both segments, in row order

    is_prefill: bool = False
    is_mixed: bool = True
