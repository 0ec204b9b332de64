from .llama import LlamaConfig, LlamaModel, PRESETS

__all__ = ["LlamaConfig", "LlamaModel", "PRESETS"]
