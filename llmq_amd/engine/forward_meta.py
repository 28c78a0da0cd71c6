"""Per-step attention metadata passed to the model forward."""

from __future__ import annotations

from dataclasses import dataclass
import torch


@dataclass
class ChunkGather:
    """Chunked-prefill context assembly: per layer, the attention K/V is the
    GATHERED past (from the paged cache) interleaved with the fresh chunk
    rows. Indices are precomputed once per step."""

    cu_seqlens_k: torch.Tensor   # [B+1] int32 key offsets (>= query lens)
    blocks: torch.Tensor         # [P] int64 cache block index per past row
    offs: torch.Tensor           # [P] int64 in-block offset per past row
    past_dst: torch.Tensor       # [P] int64 destination rows in the packed K
    fresh_dst: torch.Tensor      # [Tq] int64 destination rows in the packed K
    total_k: int                 # packed K rows


@dataclass
class PrefillMeta:
    """Packed variable-length prompt batch (whole prompts or chunks)."""

    cu_seqlens: torch.Tensor  # [B+1] int32 (query offsets)
    max_seqlen: int
    slot_mapping: torch.Tensor  # [T] int64 global KV slots
    gather: "ChunkGather | None" = None  # set when any entry is a chunk

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
