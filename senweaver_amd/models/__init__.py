from .config import ModelConfig, get_config, llama3_8b, llama3_70b, mixtral_8x7b, tiny_debug, tiny_moe, tiny_moe_tp, tiny_tp
from .io import load_weights, save_weights
from .llama import LlamaModel

__all__ = ["ModelConfig", "get_config", "llama3_8b", "llama3_70b", "mixtral_8x7b",
           "tiny_debug", "tiny_moe", "tiny_moe_tp", "tiny_tp", "LlamaModel",
           "save_weights", "load_weights"]
