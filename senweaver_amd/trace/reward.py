"""9-dimension weighted reward — semantics-identical to the reference.

Reimplements ``_computeRewardSignals`` (reference common/traceCollectorService.ts:668-788):
chatMode-adaptive thresholds, dimension order, weight table, and the exact
left-to-right weighted-sum association so finalReward values match the reference
bit-for-bit on the same inputs (all arithmetic is IEEE-754 double in both).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

from .schema import ConversationTrace, RewardDimension

# Weight table — traceCollectorService.ts:766-776; unknown dims default to 0.05 (:780)
REWARD_WEIGHTS = {
    "user_feedback": 0.25,
    "task_completion": 0.18,
    "tool_success_rate": 0.12,
    "tool_call_reliability": 0.08,
    "tool_call_efficiency": 0.05,
    "tool_duration_efficiency": 0.05,
    "response_efficiency": 0.08,
    "token_efficiency": 0.08,
    "conversation_efficiency": 0.11,
}
DEFAULT_WEIGHT = 0.05


def compute_reward_signals(trace: ConversationTrace) -> Tuple[Optional[float], List[RewardDimension]]:
    """Compute (finalReward, dims) and write them into trace.summary."""
    dims: List[RewardDimension] = []
    s = trace.summary

    chat_mode = "normal"
    if trace.metadata and isinstance(trace.metadata.get("chatMode"), str):
        chat_mode = trace.metadata["chatMode"] or "normal"
    is_agent = chat_mode == "agent"

    # Dim 1: user feedback (direct signal, highest weight)
    feedback_score = 1.0 if s.user_feedback == "good" else -1.0 if s.user_feedback == "bad" else 0.0
    dims.append(RewardDimension("user_feedback", feedback_score))

    # Dim 2: task completion
    completion = 0.5
    if trace.end_time is not None and not s.has_errors:
        completion = 0.8
    if s.has_errors:
        completion = -0.5
    if s.user_feedback == "good":
        completion = 1.0
    dims.append(RewardDimension("task_completion", completion))

    # Dims 3-5(+5b): tool-call dimensions, only when tools were called
    if s.total_tool_calls > 0:
        rate = s.tool_calls_succeeded / s.total_tool_calls
        dims.append(RewardDimension("tool_success_rate", rate * 2 - 1))

        fail_thr = {"severe": 5, "moderate": 3, "minor": 2} if is_agent else {"severe": 3, "moderate": 2, "minor": 1}
        penalty = 1.0
        if s.tool_calls_failed >= fail_thr["severe"]:
            penalty = -1.0
        elif s.tool_calls_failed >= fail_thr["moderate"]:
            penalty = -0.5
        elif s.tool_calls_failed >= fail_thr["minor"]:
            penalty = -0.2
        dims.append(RewardDimension("tool_call_reliability", penalty))

        count_thr = {"excellent": 8, "good": 15, "fair": 25} if is_agent else {"excellent": 3, "good": 6, "fair": 10}
        count_score = 1.0
        if s.total_tool_calls > count_thr["fair"]:
            count_score = -0.8
        elif s.total_tool_calls > count_thr["good"]:
            count_score = -0.3
        elif s.total_tool_calls > count_thr["excellent"]:
            count_score = 0.3
        dims.append(RewardDimension("tool_call_efficiency", count_score))

        if s.total_tool_duration_ms > 0:
            avg_dur = s.total_tool_duration_ms / s.total_tool_calls
            dur_score = 1.0
            if avg_dur > 10000:
                dur_score = -0.5
            elif avg_dur > 3000:
                dur_score = 0.0
            elif avg_dur > 1000:
                dur_score = 0.5
            dims.append(RewardDimension("tool_duration_efficiency", dur_score))

    # Dim 6: response efficiency (LLM call count)
    if s.total_llm_calls > 0:
        llm_thr = 3 if is_agent else 1
        eff = max(-1.0, 1 - max(0, s.total_llm_calls - llm_thr) * 0.4)
        dims.append(RewardDimension("response_efficiency", eff))

    # Dim 7: token efficiency
    if s.total_tokens > 0:
        tok_thr = {"excellent": 5000, "good": 15000, "fair": 30000} if is_agent else {"excellent": 2000, "good": 5000, "fair": 10000}
        tok_score = 1.0
        if s.total_tokens > tok_thr["fair"]:
            tok_score = -0.5
        elif s.total_tokens > tok_thr["good"]:
            tok_score = 0.0
        elif s.total_tokens > tok_thr["excellent"]:
            tok_score = 0.5
        dims.append(RewardDimension("token_efficiency", tok_score))

    # Dim 8: conversation depth
    user_msgs = sum(1 for sp in trace.spans if sp.type == "user_message")
    assistant_msgs = sum(1 for sp in trace.spans if sp.type == "assistant_message")
    turns = min(user_msgs, assistant_msgs)
    if turns > 0:
        turn_thr = 3 if is_agent else 2
        turn_score = 1.0
        if turns > turn_thr * 3:
            turn_score = -0.8
        elif turns > turn_thr * 2:
            turn_score = -0.3
        elif turns > turn_thr:
            turn_score = 0.3
        dims.append(RewardDimension("conversation_efficiency", turn_score))

    # Weighted composite — same association order as the reference loop (:777-784)
    weighted_sum = 0.0
    total_weight = 0.0
    for dim in dims:
        w = REWARD_WEIGHTS.get(dim.name, DEFAULT_WEIGHT)
        weighted_sum += dim.value * w
        total_weight += w
    final_reward = weighted_sum / total_weight if total_weight > 0 else None

    trace.summary.reward_dimensions = dims
    trace.summary.final_reward = final_reward
    return final_reward, dims
