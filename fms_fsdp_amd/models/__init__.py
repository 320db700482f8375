from .llama import Llama, LlamaBlock, LlamaConfig, RMSNorm

__all__ = ["Llama", "LlamaBlock", "LlamaConfig", "RMSNorm"]
