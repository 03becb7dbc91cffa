"""Model configurations for the local scorer backbones.

The reference reaches its "model" over HTTPS (20 remote providers,
sendLLMMessage.impl.ts); here the backbone is local and these configs name
the architectures BASELINE.json's configs require: Llama-3-8B (configs 2-3),
Mixtral-8x7B MoE (config 4), Llama-3-70B TP=8 (config 5), plus a tiny CPU
debug model for tests.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelConfig:
    name: str
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    vocab_size: int
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    # MoE (Mixtral-style); 0 experts = dense
    num_experts: int = 0
    num_experts_per_tok: int = 2

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


def llama3_8b() -> ModelConfig:
    return ModelConfig(
        name="llama-3-8b", hidden_size=4096, intermediate_size=14336,
        num_layers=32, num_heads=32, num_kv_heads=8, head_dim=128,
        vocab_size=128256,
    )


def llama3_70b() -> ModelConfig:
    return ModelConfig(
        name="llama-3-70b", hidden_size=8192, intermediate_size=28672,
        num_layers=80, num_heads=64, num_kv_heads=8, head_dim=128,
        vocab_size=128256,
    )


def mixtral_8x7b() -> ModelConfig:
    return ModelConfig(
        name="mixtral-8x7b", hidden_size=4096, intermediate_size=14336,
        num_layers=32, num_heads=32, num_kv_heads=8, head_dim=128,
        vocab_size=32000, rope_theta=1000000.0, num_experts=8,
        num_experts_per_tok=2,
    )


def tiny_debug(vocab: int = 512) -> ModelConfig:
    """CPU-testable model (attention D=128 to match the GPU kernel contract)."""
    return ModelConfig(
        name="tiny-debug", hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=2, num_kv_heads=1, head_dim=128,
        vocab_size=vocab, max_position=2048,
    )


def tiny_moe(vocab: int = 512) -> ModelConfig:
    """CPU-testable MoE model (grouped-GEMM path)."""
    return ModelConfig(
        name="tiny-moe", hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=2, num_kv_heads=1, head_dim=128,
        vocab_size=vocab, max_position=2048, num_experts=4,
        num_experts_per_tok=2,
    )


def tiny_moe_tp(vocab: int = 512) -> ModelConfig:
    """CPU-testable MoE model with TP-divisible heads/experts' FFN."""
    return ModelConfig(
        name="tiny-moe-tp", hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=2, num_kv_heads=2, head_dim=128,
        vocab_size=vocab, max_position=2048, num_experts=4,
        num_experts_per_tok=2,
    )


def tiny_tp(vocab: int = 512) -> ModelConfig:
    """CPU-testable TP model (heads divisible by 2)."""
    return ModelConfig(
        name="tiny-tp", hidden_size=512, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=128,
        vocab_size=vocab, max_position=2048,
    )


PRESETS = {
    "llama-3-8b": llama3_8b,
    "llama-3-70b": llama3_70b,
    "mixtral-8x7b": mixtral_8x7b,
    "tiny-debug": tiny_debug,
    "tiny-moe": tiny_moe,
    "tiny-tp": tiny_tp,
}


def get_config(name: str) -> ModelConfig:
    if name not in PRESETS:
        raise KeyError(f"unknown model preset {name!r}; have {sorted(PRESETS)}")
    return PRESETS[name]()
