"""Trace span / conversation-trace schema.

Format-compatible with the reference's TraceSpan / ConversationTrace interfaces
(reference: common/traceCollectorService.ts:40-109).  JSON uses camelCase keys;
fields the reference leaves ``undefined`` are omitted from serialized output.
"""

from __future__ import annotations

import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

SPAN_TYPES = (
    "llm_call",
    "tool_call",
    "user_message",
    "assistant_message",
    "user_feedback",
    "edit_prediction",
    "checkpoint",
    "error",
)

# Caps — identical to reference common/traceCollectorService.ts:218-221
MAX_CONTENT_PREVIEW = 500
MAX_TRACES = 1000
MAX_SPANS_PER_TRACE = 200
FLUSH_INTERVAL_MS = 30000


def new_uuid() -> str:
    return str(uuid.uuid4())


def truncate(s: Optional[str], max_len: int = MAX_CONTENT_PREVIEW) -> str:
    """Reference _truncate (traceCollectorService.ts:259-262): '' for falsy,
    first max_len chars + '...' when longer."""
    if not s:
        return ""
    return s[:max_len] + "..." if len(s) > max_len else s


@dataclass
class RewardDimension:
    name: str
    value: float

    def to_json(self) -> Dict[str, Any]:
        return {"name": self.name, "value": self.value}

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "RewardDimension":
        return cls(name=d["name"], value=d["value"])


@dataclass
class TraceSpan:
    id: str
    trace_id: str
    thread_id: str
    message_idx: int
    type: str
    timestamp: int
    data: Dict[str, Any] = field(default_factory=dict)
    duration: Optional[float] = None

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "id": self.id,
            "traceId": self.trace_id,
            "threadId": self.thread_id,
            "messageIdx": self.message_idx,
            "type": self.type,
            "timestamp": self.timestamp,
        }
        if self.duration is not None:
            out["duration"] = self.duration
        # data keys with undefined/None values are omitted (JSON.stringify behavior)
        out["data"] = {k: v for k, v in self.data.items() if v is not None or k == "feedback"}
        return out

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "TraceSpan":
        return cls(
            id=d["id"],
            trace_id=d.get("traceId", ""),
            thread_id=d.get("threadId", ""),
            message_idx=d.get("messageIdx", 0),
            type=d["type"],
            timestamp=d.get("timestamp", 0),
            data=dict(d.get("data", {})),
            duration=d.get("duration"),
        )


@dataclass
class TraceSummary:
    """Aggregated info — reference ConversationTrace.summary (:95-108)."""

    total_llm_calls: int = 0
    total_tool_calls: int = 0
    total_tokens: int = 0
    user_feedback: Optional[str] = None  # 'good' | 'bad' | None
    has_errors: bool = False
    tool_calls_succeeded: int = 0
    tool_calls_failed: int = 0
    tool_calls_by_name: Dict[str, Dict[str, int]] = field(default_factory=dict)
    total_tool_duration_ms: float = 0
    final_reward: Optional[float] = None
    reward_dimensions: List[RewardDimension] = field(default_factory=list)

    def to_json(self) -> Dict[str, Any]:
        return {
            "totalLLMCalls": self.total_llm_calls,
            "totalToolCalls": self.total_tool_calls,
            "totalTokens": self.total_tokens,
            "userFeedback": self.user_feedback,
            "hasErrors": self.has_errors,
            "toolCallsSucceeded": self.tool_calls_succeeded,
            "toolCallsFailed": self.tool_calls_failed,
            "toolCallsByName": self.tool_calls_by_name,
            "totalToolDurationMs": self.total_tool_duration_ms,
            "finalReward": self.final_reward,
            "rewardDimensions": [d.to_json() for d in self.reward_dimensions],
        }

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "TraceSummary":
        return cls(
            total_llm_calls=d.get("totalLLMCalls", 0),
            total_tool_calls=d.get("totalToolCalls", 0),
            total_tokens=d.get("totalTokens", 0),
            user_feedback=d.get("userFeedback"),
            has_errors=d.get("hasErrors", False),
            tool_calls_succeeded=d.get("toolCallsSucceeded", 0),
            tool_calls_failed=d.get("toolCallsFailed", 0),
            tool_calls_by_name={k: dict(v) for k, v in d.get("toolCallsByName", {}).items()},
            total_tool_duration_ms=d.get("totalToolDurationMs", 0),
            final_reward=d.get("finalReward"),
            reward_dimensions=[RewardDimension.from_json(x) for x in d.get("rewardDimensions", [])],
        )


@dataclass
class ConversationTrace:
    id: str
    thread_id: str
    start_time: int
    spans: List[TraceSpan] = field(default_factory=list)
    end_time: Optional[int] = None
    metadata: Optional[Dict[str, Any]] = None
    summary: TraceSummary = field(default_factory=TraceSummary)

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "id": self.id,
            "threadId": self.thread_id,
            "startTime": self.start_time,
        }
        if self.end_time is not None:
            out["endTime"] = self.end_time
        out["spans"] = [s.to_json() for s in self.spans]
        if self.metadata is not None:
            out["metadata"] = self.metadata
        out["summary"] = self.summary.to_json()
        return out

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "ConversationTrace":
        return cls(
            id=d["id"],
            thread_id=d.get("threadId", ""),
            start_time=d.get("startTime", 0),
            spans=[TraceSpan.from_json(s) for s in d.get("spans", [])],
            end_time=d.get("endTime"),
            metadata=d.get("metadata"),
            summary=TraceSummary.from_json(d.get("summary", {})),
        )
