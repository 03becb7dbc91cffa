from .schema import (
    ConversationTrace,
    RewardDimension,
    TraceSpan,
    TraceSummary,
    MAX_CONTENT_PREVIEW,
    MAX_SPANS_PER_TRACE,
    MAX_TRACES,
    truncate,
)
from .reward import compute_reward_signals, REWARD_WEIGHTS
from .collector import TraceCollector

__all__ = [
    "ConversationTrace",
    "RewardDimension",
    "TraceSpan",
    "TraceSummary",
    "TraceCollector",
    "compute_reward_signals",
    "REWARD_WEIGHTS",
    "MAX_CONTENT_PREVIEW",
    "MAX_SPANS_PER_TRACE",
    "MAX_TRACES",
    "truncate",
]
