from .kvcache import PagedKVCache
from .engine import LlamaEngine, Sequence, GenParams

__all__ = ["PagedKVCache", "LlamaEngine", "Sequence", "GenParams"]
