"""6-pattern failure detector + reward-dimension patterns.

Semantics-identical to the reference's _analyzePatterns (apoService.ts:635-773)
and the reward-dimension pattern pass (:574-596).  Thresholds verbatim:

  P1 errors->bad          >=2 traces (high at >=5)
  P2 tool-fail->bad       >=2        (high at >=5)
  P3 tokens>10000->bad    >=3        (always medium)
  P4 LLM calls>2->bad     >=2        (always high)
  P5 user msgs>=4->bad    >=2        (high at >=4)
  P6 tool duration>15s->bad >=2      (always medium)
  Reward-dim: avg < -0.3 and n>=5    (high at avg < -0.5)
"""

from __future__ import annotations

from typing import Callable, Dict, List

from ..trace.schema import ConversationTrace
from ..utils.jsonutil import to_fixed
from .schema import PatternExample, PromptIssuePattern

# dimension -> category map (apoService.ts:576-586 and :803-813)
DIMENSION_CATEGORY_MAP = {
    "tool_success_rate": "tool_usage",
    "tool_call_reliability": "tool_usage",
    "tool_call_efficiency": "tool_usage",
    "tool_duration_efficiency": "tool_usage",
    "token_efficiency": "context_management",
    "response_efficiency": "core_behavior",
    "conversation_efficiency": "core_behavior",
    "task_completion": "core_behavior",
    "user_feedback": "core_behavior",
}


def _first_span_preview(trace: ConversationTrace, span_type: str) -> str:
    for sp in trace.spans:
        if sp.type == span_type:
            return sp.data.get("contentPreview") or ""
    return ""


def _examples_user_assistant(traces: List[ConversationTrace]) -> List[PatternExample]:
    out = []
    for t in traces[:3]:
        out.append(PatternExample(
            thread_id=t.thread_id,
            user_message_preview=_first_span_preview(t, "user_message"),
            assistant_message_preview=_first_span_preview(t, "assistant_message"),
            feedback=t.summary.user_feedback,
        ))
    return out


def _examples_with_assistant_text(traces: List[ConversationTrace], text_fn: Callable[[ConversationTrace], str]) -> List[PatternExample]:
    out = []
    for t in traces[:3]:
        out.append(PatternExample(
            thread_id=t.thread_id,
            user_message_preview=_first_span_preview(t, "user_message"),
            assistant_message_preview=text_fn(t),
            feedback=t.summary.user_feedback,
        ))
    return out


def analyze_patterns(traces: List[ConversationTrace], uuid_fn: Callable[[], str]) -> List[PromptIssuePattern]:
    """The 6 trace-level patterns, evaluated over bad-feedback traces."""
    patterns: List[PromptIssuePattern] = []
    bad_examples = [t for t in traces if t.summary.user_feedback == "bad"]
    if not bad_examples:
        return patterns

    # Pattern 1: errors -> bad feedback
    error_traces = [t for t in traces if t.summary.has_errors and t.summary.user_feedback == "bad"]
    if len(error_traces) >= 2:
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="Users give negative feedback after errors occur in conversations",
            frequency=len(error_traces),
            severity="high" if len(error_traces) >= 5 else "medium",
            related_category="core_behavior",
            examples=_examples_user_assistant(error_traces),
        ))

    # Pattern 2: tool-call failures -> bad feedback
    def _has_failed_tool(t: ConversationTrace) -> bool:
        return any(sp.type == "tool_call" and sp.data.get("toolSuccess") is False for sp in t.spans)

    tool_fail_traces = [t for t in traces if _has_failed_tool(t) and t.summary.user_feedback == "bad"]
    if len(tool_fail_traces) >= 2:
        def _failed_tool_text(t: ConversationTrace) -> str:
            for sp in t.spans:
                if sp.type == "tool_call" and sp.data.get("toolSuccess") is False:
                    result = (sp.data.get("toolResult") or "")[:100]
                    return f"Tool {sp.data.get('toolName')} failed: {result}"
            return "Tool undefined failed: "
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="Tool call failures lead to user dissatisfaction",
            frequency=len(tool_fail_traces),
            severity="high" if len(tool_fail_traces) >= 5 else "medium",
            related_category="tool_usage",
            examples=_examples_with_assistant_text(tool_fail_traces, _failed_tool_text),
        ))

    # Pattern 3: high token consumption -> bad feedback
    high_token = [t for t in traces if t.summary.total_tokens > 10000 and t.summary.user_feedback == "bad"]
    if len(high_token) >= 3:
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="User feedback is poor in conversations with high token consumption",
            frequency=len(high_token),
            severity="medium",
            related_category="context_management",
            examples=_examples_with_assistant_text(high_token, lambda t: f"Total tokens: {t.summary.total_tokens}"),
        ))

    # Pattern 4: multiple LLM calls, still bad
    multi_call = [t for t in traces if t.summary.total_llm_calls > 2 and t.summary.user_feedback == "bad"]
    if len(multi_call) >= 2:
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="Users still dissatisfied after multiple LLM calls (possible retries)",
            frequency=len(multi_call),
            severity="high",
            related_category="core_behavior",
            examples=_examples_with_assistant_text(multi_call, lambda t: f"LLM calls: {t.summary.total_llm_calls}"),
        ))

    # Pattern 5: long conversations (>=4 user msgs), still bad
    def _user_msg_count(t: ConversationTrace) -> int:
        return sum(1 for sp in t.spans if sp.type == "user_message")

    long_conv = [t for t in traces if _user_msg_count(t) >= 4 and t.summary.user_feedback == "bad"]
    if len(long_conv) >= 2:
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="Long conversations with many turns still result in user dissatisfaction",
            frequency=len(long_conv),
            severity="high" if len(long_conv) >= 4 else "medium",
            related_category="core_behavior",
            examples=_examples_with_assistant_text(long_conv, lambda t: f"Conversation turns: {_user_msg_count(t)}"),
        ))

    # Pattern 6: slow tool calls (>15s total), bad
    slow_tool = [t for t in traces if t.summary.total_tool_duration_ms > 15000 and t.summary.user_feedback == "bad"]
    if len(slow_tool) >= 2:
        patterns.append(PromptIssuePattern(
            id=uuid_fn(),
            description="Slow tool execution (>15s total) correlates with user dissatisfaction",
            frequency=len(slow_tool),
            severity="medium",
            related_category="tool_usage",
            examples=_examples_with_assistant_text(
                slow_tool, lambda t: f"Tool duration: {to_fixed(t.summary.total_tool_duration_ms / 1000, 1)}s"),
        ))

    return patterns


def reward_dimension_patterns(
    reward_by_dimension: Dict[str, Dict[str, float]], uuid_fn: Callable[[], str]
) -> List[PromptIssuePattern]:
    """Patterns from consistently-low reward dimensions (apoService.ts:574-596)."""
    patterns: List[PromptIssuePattern] = []
    for dim_name, st in reward_by_dimension.items():
        if st["avg"] < -0.3 and st["count"] >= 5:
            patterns.append(PromptIssuePattern(
                id=uuid_fn(),
                description=f"{dim_name} dimension reward signal consistently low (avg: {to_fixed(st['avg'], 3)})",
                frequency=int(st["count"]),
                severity="high" if st["avg"] < -0.5 else "medium",
                related_category=DIMENSION_CATEGORY_MAP.get(dim_name, "core_behavior"),
                examples=[],
            ))
    return patterns
