from .kvcache import PagedKVCache, PAGE_SIZE
from .tokenizer import BPETokenizer, HashTokenizer
from .scorer import LlamaBackend

__all__ = ["PagedKVCache", "PAGE_SIZE", "BPETokenizer", "HashTokenizer", "LlamaBackend"]
