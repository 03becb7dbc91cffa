"""Model capability registry (modelCapabilities.ts rebuild, local edition).

The reference keeps a 2211-LoC static table mapping (provider, model) to
capability records — contextWindow, reservedOutputTokenSpace,
supportsSystemMessage / specialToolFormat / supportsFIM / cost /
reasoningCapabilities — consumed by the context-fitting pipeline and the
send path (getModelCapabilities :2108, getReservedOutputTokenSpace :2177).
Here the "providers" are the local backbones, so the registry covers the
engine's presets with the same field algebra plus the same lookup
semantics: unknown model names resolve through recognizable-substring
matching to a default record, exactly like the reference's fallback.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

DEFAULT_RESERVED_OUTPUT_TOKEN_SPACE = 4096


@dataclass
class ReasoningCapabilities:
    supportsReasoning: bool = False
    canTurnOffReasoning: bool = True
    canIOReasoning: bool = True
    reasoningReservedOutputTokenSpace: Optional[int] = None
    openSourceThinkTags: Optional[tuple] = ("<think>", "</think>")


@dataclass
class ModelCapabilities:
    contextWindow: int
    reservedOutputTokenSpace: Optional[int] = None  # None -> default 4096
    supportsSystemMessage: object = "system-role"   # False | 'system-role' |
    #                                    'developer-role' | 'separated'
    specialToolFormat: Optional[str] = None  # None => XML tool calls in agent mode
    supportsFIM: bool = False
    reasoningCapabilities: object = False   # False | ReasoningCapabilities
    maxOutputTokens: Optional[int] = None
    downloadable: bool = False


_LOCAL = "senweaver_amd"  # the single local "provider"

# per-preset capability records (the engine's XML tool grammar means
# specialToolFormat stays None; the local models are instruction-tuned
# Llama/Mixtral-family shapes)
_MODEL_CAPS = {
    "llama-3-8b": ModelCapabilities(contextWindow=8192,
                                    reservedOutputTokenSpace=2048,
                                    supportsFIM=True),
    "llama-3-70b": ModelCapabilities(contextWindow=8192,
                                     reservedOutputTokenSpace=2048),
    "mixtral-8x7b": ModelCapabilities(contextWindow=32768,
                                      reservedOutputTokenSpace=4096),
    "tiny-debug": ModelCapabilities(contextWindow=2048,
                                    reservedOutputTokenSpace=256),
    "tiny-moe": ModelCapabilities(contextWindow=2048,
                                  reservedOutputTokenSpace=256),
    "tiny-moe-tp": ModelCapabilities(contextWindow=2048,
                                     reservedOutputTokenSpace=256),
    "tiny-tp": ModelCapabilities(contextWindow=2048,
                                 reservedOutputTokenSpace=256),
}

# recognizable-substring fallbacks (the reference resolves e.g. any
# "llama"-named model to the llama family record)
_FAMILY_SUBSTRINGS = [
    ("llama", _MODEL_CAPS["llama-3-8b"]),
    ("mixtral", _MODEL_CAPS["mixtral-8x7b"]),
    ("tiny", _MODEL_CAPS["tiny-debug"]),
]

_DEFAULT = ModelCapabilities(contextWindow=4096)


def get_model_capabilities(model_name: str,
                           provider: str = _LOCAL) -> ModelCapabilities:
    if provider != _LOCAL:
        return _DEFAULT
    caps = _MODEL_CAPS.get(model_name)
    if caps is not None:
        return caps
    low = model_name.lower()
    for sub, rec in _FAMILY_SUBSTRINGS:
        if sub in low:
            return rec
    return _DEFAULT


def get_reserved_output_token_space(model_name: str, *, is_reasoning: bool = False,
                                    provider: str = _LOCAL) -> int:
    caps = get_model_capabilities(model_name, provider)
    if is_reasoning and isinstance(caps.reasoningCapabilities,
                                   ReasoningCapabilities):
        override = caps.reasoningCapabilities.reasoningReservedOutputTokenSpace
        if override is not None:
            return override
    if caps.reservedOutputTokenSpace is not None:
        return caps.reservedOutputTokenSpace
    return DEFAULT_RESERVED_OUTPUT_TOKEN_SPACE


def get_context_window(model_name: str, provider: str = _LOCAL) -> int:
    return get_model_capabilities(model_name, provider).contextWindow
