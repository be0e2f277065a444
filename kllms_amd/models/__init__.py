from .llama import LlamaForCausalLM, ForwardBatch

__all__ = ["LlamaForCausalLM", "ForwardBatch"]
