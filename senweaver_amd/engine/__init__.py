from .kvcache import PagedKVCache, PAGE_SIZE
from .tokenizer import HashTokenizer
from .scorer import LlamaBackend

__all__ = ["PagedKVCache", "PAGE_SIZE", "HashTokenizer", "LlamaBackend"]
