from .llama import (AttnMetadata, LLAMA_CONFIGS, LlamaConfig, LlamaForCausalLM)

__all__ = ["AttnMetadata", "LLAMA_CONFIGS", "LlamaConfig", "LlamaForCausalLM"]
