"""In-tree MI355X inference engine.

Replaces the reference's vLLM dependency (consumed surface documented in
SURVEY §2.9: AsyncEngineArgs / AsyncLLMEngine.from_engine_args /
engine.generate / engine.get_tokenizer, vllm_worker.py:105-186).

Components:
  EngineConfig      engine knobs (max_num_seqs, gpu_memory_utilization, ...)
  SamplingParams    per-request sampling settings
  LLMEngine         synchronous continuous-batching engine (step loop)
  AsyncEngine       asyncio facade: generate() async streams, engine thread
"""

from llmq_amd.engine.config import EngineConfig
from llmq_amd.engine.sampling_params import SamplingParams

__all__ = ["EngineConfig", "SamplingParams"]
