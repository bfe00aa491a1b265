from .llama import LlamaConfig, LlamaForCausalLM, CONFIGS

__all__ = ["LlamaConfig", "LlamaForCausalLM", "CONFIGS"]
