from .llama import LlamaForCausalLM

__all__ = ["LlamaForCausalLM"]
