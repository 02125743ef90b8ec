"""Model families for the on-node engine (Llama-architecture decoders)."""

from .config import LlamaConfig, PRESETS, get_config
from .llama import LlamaModel, PagedKVCache

__all__ = ["LlamaConfig", "PRESETS", "get_config", "LlamaModel", "PagedKVCache"]
