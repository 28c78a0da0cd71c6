from llmq_amd.engine.models.llama import CausalLM

__all__ = ["CausalLM"]
