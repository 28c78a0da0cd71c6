"""Engine configuration (the AsyncEngineArgs analogue)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch


@dataclass
class EngineConfig:
    model: str = "tiny-llama"
    dtype: str = "bfloat16"  # compute dtype on GPU; float32 on CPU
    max_num_seqs: int = 256
    max_model_len: Optional[int] = None
    max_prefill_tokens: int = 8192  # token budget per prefill step
    gpu_memory_utilization: float = 0.9
    kv_block_size: int = 16
    # "auto" = model dtype; "fp8" = OCP e4m3 KV storage (bf16 compute) —
    # halves the decode KV stream; opt-in, off by default.
    kv_cache_dtype: str = "auto"
    enable_hipgraph: bool = True
    hipgraph_max_batch: int = 512
    tensor_parallel_size: int = 1
    seed: int = 0
    device: str = "auto"  # "auto" | "cuda" | "cpu"
    load_weights: bool = True  # False → random init (synthetic benchmarking)
    fast_init: bool = False  # device-side random init (large models)
    # CPU-test override: number of KV blocks (None → sized from free HBM)
    num_kv_blocks: Optional[int] = None
    enforce_eager: bool = False

    def resolve_device(self) -> torch.device:
        if self.device == "auto":
            return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        return torch.device(self.device)

    def resolve_kv_dtype(self, device: torch.device) -> torch.dtype:
        model_dtype = self.resolve_dtype(device)
        if self.kv_cache_dtype == "auto":
            return model_dtype
        if self.kv_cache_dtype == "fp8":
            if device.type != "cuda" or model_dtype != torch.bfloat16:
                raise ValueError("kv_cache_dtype=fp8 requires GPU bf16 execution")
            return torch.float8_e4m3fn
        raise ValueError(f"unknown kv_cache_dtype {self.kv_cache_dtype!r}")

    def resolve_dtype(self, device: torch.device) -> torch.dtype:
        if device.type == "cpu":
            return torch.float32
        return {
            "bfloat16": torch.bfloat16,
            "bf16": torch.bfloat16,
            "float16": torch.float16,
            "fp16": torch.float16,
            "float32": torch.float32,
        }[self.dtype]
