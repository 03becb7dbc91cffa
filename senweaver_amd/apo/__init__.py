from .schema import (
    APO_RULES_MAX_CHARS,
    BeamSearchState,
    MAX_GRADIENTS,
    MAX_REPORTS,
    MAX_SUGGESTIONS,
    PromptEffectivenessReport,
    PromptIssuePattern,
    PromptOptimizationSuggestion,
    PromptSegment,
    RolloutMessage,
    RolloutResult,
    TextualGradient,
    VersionedPromptTemplate,
    default_apo_config,
)
from .patterns import analyze_patterns, reward_dimension_patterns, DIMENSION_CATEGORY_MAP
from .prompts import build_apply_edit_prompt, build_textual_gradient_prompt
from .rules import inject_rules, pack_rules
from .service import APOService
from .optimizer import LocalGradientEngine, PromptOptimizerBackend, StubBackend, rollout_weight
from .beam import BeamSearchEngine

__all__ = [
    "APOService",
    "APO_RULES_MAX_CHARS",
    "BeamSearchEngine",
    "BeamSearchState",
    "DIMENSION_CATEGORY_MAP",
    "LocalGradientEngine",
    "MAX_GRADIENTS",
    "MAX_REPORTS",
    "MAX_SUGGESTIONS",
    "PromptEffectivenessReport",
    "PromptIssuePattern",
    "PromptOptimizationSuggestion",
    "PromptOptimizerBackend",
    "PromptSegment",
    "RolloutMessage",
    "RolloutResult",
    "StubBackend",
    "TextualGradient",
    "VersionedPromptTemplate",
    "analyze_patterns",
    "build_apply_edit_prompt",
    "build_textual_gradient_prompt",
    "default_apo_config",
    "inject_rules",
    "pack_rules",
    "reward_dimension_patterns",
    "rollout_weight",
]
