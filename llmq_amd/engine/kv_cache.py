"""Paged KV cache + block allocator.

Layout per layer: [num_blocks, num_kv_heads, block_size, head_dim] — a
(block, kv_head) pair is a contiguous [block_size, head_dim] tile, which is
what one workgroup of the CDNA4 decode kernel streams with coalesced
dwordx4 loads (PagedAttention equivalent).

Sizing: the engine claims ``gpu_memory_utilization`` × free HBM after
weights are resident (MI355X: 288 GB per GPU, so 9B-bf16 leaves ~240 GB of
KV — tens of thousands of 8k contexts; the scheduler, not memory, is the
usual admission limit)."""

from __future__ import annotations

import logging
from typing import List, Optional

import torch

logger = logging.getLogger(__name__)


class KVCache:
    def __init__(
        self,
        num_layers: int,
        num_blocks: int,
        num_kv_heads: int,
        block_size: int,
        head_dim: int,
        device: torch.device,
        dtype: torch.dtype,
    ):
        self.num_layers = num_layers
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.k: List[torch.Tensor] = []
        self.v: List[torch.Tensor] = []
        for _ in range(num_layers):
            self.k.append(
                torch.zeros(
                    num_blocks, num_kv_heads, block_size, head_dim, device=device, dtype=dtype
                )
            )
            self.v.append(
                torch.zeros(
                    num_blocks, num_kv_heads, block_size, head_dim, device=device, dtype=dtype
                )
            )

    @staticmethod
    def block_bytes(num_layers, num_kv_heads, block_size, head_dim, dtype) -> int:
        elem = torch.tensor([], dtype=dtype).element_size()
        return 2 * num_layers * num_kv_heads * block_size * head_dim * elem

    @classmethod
    def num_blocks_for_budget(
        cls, budget_bytes: int, num_layers, num_kv_heads, block_size, head_dim, dtype
    ) -> int:
        per_block = cls.block_bytes(num_layers, num_kv_heads, block_size, head_dim, dtype)
        return max(budget_bytes // per_block, 0)


class BlockAllocator:
    """Free-list allocator over block ids [0, num_blocks)."""

    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int) -> bool:
        return len(self._free) >= n

    def allocate(self, n: int) -> Optional[List[int]]:
        if len(self._free) < n:
            return None
        if n == 0:
            return []
        blocks = self._free[-n:][::-1]
        del self._free[-n:]
        return blocks

    def free(self, blocks: List[int]) -> None:
        self._free.extend(reversed(blocks))


def blocks_needed(num_tokens: int, block_size: int) -> int:
    return (num_tokens + block_size - 1) // block_size
