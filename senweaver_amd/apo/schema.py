"""APO schema — format-compatible with the reference's apoService.ts types.

Reference: common/apoService.ts:22-200 (PromptSegmentCategory, PromptSegment,
PromptEffectivenessReport, PromptIssuePattern, PromptOptimizationSuggestion,
RolloutResultForAPO, VersionedPromptTemplate, TextualGradient, BeamSearchState,
APOConfig).  JSON is camelCase; ``undefined`` fields are omitted.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

SEGMENT_CATEGORIES = (
    "core_behavior",
    "code_quality",
    "tool_usage",
    "output_format",
    "context_management",
    "mode_specific",
    "user_instructions",
)

# Caps — apoService.ts:276-277, :405
MAX_REPORTS = 50
MAX_SUGGESTIONS = 200
MAX_GRADIENTS = 50

# Rule-injection budget — browser/convertToLLMMessageService.ts:835
APO_RULES_MAX_CHARS = 2000


def default_apo_config() -> Dict[str, Any]:
    """DEFAULT_APO_CONFIG — apoService.ts:279-292 (same keys/values)."""
    return {
        "enabled": True,
        "autoAnalyzeEnabled": True,
        "autoAnalyzeIntervalMs": 3600000,
        "minTracesForAnalysis": 20,
        "minFeedbacksForAnalysis": 10,
        "autoApplySuggestions": False,
        "uploadOptimizationsToServer": True,
        "beamWidth": 4,
        "branchFactor": 4,
        "beamRounds": 3,
        "gradientBatchSize": 4,
    }


@dataclass
class PromptSegment:
    id: str
    category: str
    content: str
    is_active: bool
    is_optimized: bool
    version: int
    created_at: int
    updated_at: int
    original_content: Optional[str] = None

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "id": self.id,
            "category": self.category,
            "content": self.content,
            "isActive": self.is_active,
            "isOptimized": self.is_optimized,
        }
        if self.original_content is not None:
            out["originalContent"] = self.original_content
        out.update({
            "version": self.version,
            "createdAt": self.created_at,
            "updatedAt": self.updated_at,
        })
        return out

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "PromptSegment":
        return cls(
            id=d["id"], category=d["category"], content=d["content"],
            is_active=d.get("isActive", True), is_optimized=d.get("isOptimized", False),
            version=d.get("version", 1), created_at=d.get("createdAt", 0),
            updated_at=d.get("updatedAt", 0), original_content=d.get("originalContent"),
        )


@dataclass
class PatternExample:
    thread_id: str
    user_message_preview: str
    assistant_message_preview: str
    feedback: Optional[str]

    def to_json(self) -> Dict[str, Any]:
        return {
            "threadId": self.thread_id,
            "userMessagePreview": self.user_message_preview,
            "assistantMessagePreview": self.assistant_message_preview,
            "feedback": self.feedback,
        }


@dataclass
class PromptIssuePattern:
    id: str
    description: str
    frequency: int
    severity: str  # 'low' | 'medium' | 'high'
    related_category: str
    examples: List[PatternExample] = field(default_factory=list)

    def to_json(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "description": self.description,
            "frequency": self.frequency,
            "severity": self.severity,
            "relatedCategory": self.related_category,
            "examples": [e.to_json() for e in self.examples],
        }


@dataclass
class PromptOptimizationSuggestion:
    id: str
    target_category: str
    type: str  # 'add' | 'modify' | 'remove' | 'reorder'
    priority: str  # 'low' | 'medium' | 'high'
    description: str
    reasoning: str
    estimated_impact: str
    status: str = "pending"  # 'pending' | 'applied' | 'rejected' | 'reverted'
    target_segment_id: Optional[str] = None
    suggested_content: Optional[str] = None
    applied_at: Optional[int] = None
    prompt_version: Optional[str] = None
    validation_score: Optional[float] = None

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"id": self.id, "targetCategory": self.target_category}
        if self.target_segment_id is not None:
            out["targetSegmentId"] = self.target_segment_id
        out.update({
            "type": self.type,
            "priority": self.priority,
            "description": self.description,
        })
        if self.suggested_content is not None:
            out["suggestedContent"] = self.suggested_content
        out.update({
            "reasoning": self.reasoning,
            "estimatedImpact": self.estimated_impact,
            "status": self.status,
        })
        if self.applied_at is not None:
            out["appliedAt"] = self.applied_at
        if self.prompt_version is not None:
            out["promptVersion"] = self.prompt_version
        if self.validation_score is not None:
            out["validationScore"] = self.validation_score
        return out

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "PromptOptimizationSuggestion":
        return cls(
            id=d["id"], target_category=d.get("targetCategory", "core_behavior"),
            type=d.get("type", "modify"), priority=d.get("priority", "medium"),
            description=d.get("description", ""), reasoning=d.get("reasoning", ""),
            estimated_impact=d.get("estimatedImpact", ""), status=d.get("status", "pending"),
            target_segment_id=d.get("targetSegmentId"), suggested_content=d.get("suggestedContent"),
            applied_at=d.get("appliedAt"), prompt_version=d.get("promptVersion"),
            validation_score=d.get("validationScore"),
        )


@dataclass
class PromptEffectivenessReport:
    id: str
    generated_at: int
    period: Dict[str, int]
    total_conversations: int
    good_feedback_count: int
    bad_feedback_count: int
    no_feedback_count: int
    good_rate: float
    by_mode: Dict[str, Dict[str, float]]
    patterns: List[PromptIssuePattern]
    suggestions: List[PromptOptimizationSuggestion]

    def to_json(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "generatedAt": self.generated_at,
            "period": self.period,
            "totalConversations": self.total_conversations,
            "goodFeedbackCount": self.good_feedback_count,
            "badFeedbackCount": self.bad_feedback_count,
            "noFeedbackCount": self.no_feedback_count,
            "goodRate": self.good_rate,
            "byMode": self.by_mode,
            "patterns": [p.to_json() for p in self.patterns],
            "suggestions": [s.to_json() for s in self.suggestions],
        }


@dataclass
class VersionedPromptTemplate:
    version: str
    content: str
    score: Optional[float]
    created_at: int
    parent_version: Optional[str] = None

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "version": self.version,
            "content": self.content,
            "score": self.score,
        }
        if self.parent_version is not None:
            out["parentVersion"] = self.parent_version
        out["createdAt"] = self.created_at
        return out

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "VersionedPromptTemplate":
        return cls(
            version=d["version"], content=d.get("content", ""), score=d.get("score"),
            created_at=d.get("createdAt", 0), parent_version=d.get("parentVersion"),
        )


@dataclass
class TextualGradient:
    id: str
    prompt_version: str
    critique: str
    rollout_summary: str
    created_at: int

    def to_json(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "promptVersion": self.prompt_version,
            "critique": self.critique,
            "rolloutSummary": self.rollout_summary,
            "createdAt": self.created_at,
        }

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "TextualGradient":
        return cls(
            id=d["id"], prompt_version=d.get("promptVersion", "v0"),
            critique=d.get("critique", ""), rollout_summary=d.get("rolloutSummary", ""),
            created_at=d.get("createdAt", 0),
        )


@dataclass
class BeamSearchState:
    current_round: int
    total_rounds: int
    beam: List[VersionedPromptTemplate]
    history_best_prompt: Optional[VersionedPromptTemplate]
    history_best_score: float
    version_counter: int
    started_at: int
    last_updated_at: int

    def to_json(self) -> Dict[str, Any]:
        return {
            "currentRound": self.current_round,
            "totalRounds": self.total_rounds,
            "beam": [b.to_json() for b in self.beam],
            "historyBestPrompt": self.history_best_prompt.to_json() if self.history_best_prompt else None,
            "historyBestScore": self.history_best_score,
            "versionCounter": self.version_counter,
            "startedAt": self.started_at,
            "lastUpdatedAt": self.last_updated_at,
        }

    @classmethod
    def from_json(cls, d: Dict[str, Any]) -> "BeamSearchState":
        best = d.get("historyBestPrompt")
        return cls(
            current_round=d.get("currentRound", 0),
            total_rounds=d.get("totalRounds", 3),
            beam=[VersionedPromptTemplate.from_json(b) for b in d.get("beam", [])],
            history_best_prompt=VersionedPromptTemplate.from_json(best) if best else None,
            history_best_score=d.get("historyBestScore", float("-inf")) if d.get("historyBestScore") is not None else float("-inf"),
            version_counter=d.get("versionCounter", 0),
            started_at=d.get("startedAt", 0),
            last_updated_at=d.get("lastUpdatedAt", 0),
        )


@dataclass
class RolloutMessage:
    role: str  # 'user' | 'assistant' | 'tool'
    content: str
    tool_name: Optional[str] = None
    tool_success: Optional[bool] = None

    def to_json(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"role": self.role, "content": self.content}
        if self.tool_name is not None:
            out["toolName"] = self.tool_name
        if self.tool_success is not None:
            out["toolSuccess"] = self.tool_success
        return out


@dataclass
class RolloutResult:
    """RolloutResultForAPO — apoService.ts:108-135."""

    trace_id: str
    thread_id: str
    status: str  # 'succeeded' | 'failed' | 'unknown'
    final_reward: Optional[float]
    reward_dimensions: List[Dict[str, Any]]
    messages: List[RolloutMessage]
    chat_mode: str
    tool_call_stats: Dict[str, Any]
    llm_stats: Dict[str, Any]

    def to_json(self) -> Dict[str, Any]:
        return {
            "traceId": self.trace_id,
            "threadId": self.thread_id,
            "status": self.status,
            "finalReward": self.final_reward,
            "rewardDimensions": self.reward_dimensions,
            "messages": [m.to_json() for m in self.messages],
            "chatMode": self.chat_mode,
            "toolCallStats": self.tool_call_stats,
            "llmStats": self.llm_stats,
        }
