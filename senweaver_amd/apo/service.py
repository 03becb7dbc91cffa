"""APOService — effectiveness analysis, suggestion lifecycle, beam/gradient state.

Capability- and format-compatible rebuild of the reference's APOService
(reference: common/apoService.ts).  The reference delegates the LLM work —
textual-gradient critique and beam-candidate scoring — to its backend over
HTTPS (POST /api/apo/gradient, /api/apo/optimize, apoService.ts:992-1343);
here the same requests are served by a *local optimizer* (``senweaver_amd.apo
.gradient.LocalGradientEngine`` / ``senweaver_amd.apo.beam.BeamSearchEngine``)
running the Llama backbone on MI355X HIP kernels.  Payload shapes, beam-state
update rules and suggestion/segment lifecycle are semantics-identical.
"""

from __future__ import annotations

import time
from typing import Any, Callable, Dict, List, Optional

from ..storage import (
    APO_BEAM_STATE_KEY,
    APO_CONFIG_KEY,
    APO_GRADIENTS_KEY,
    APO_SEGMENTS_KEY,
    APO_STORAGE_KEY,
    MemoryStorage,
)
from ..trace.collector import TraceCollector
from ..trace.schema import ConversationTrace, new_uuid
from ..utils.jsonutil import js_parse, js_stringify, to_fixed
from .patterns import DIMENSION_CATEGORY_MAP, analyze_patterns, reward_dimension_patterns
from .prompts import build_apply_edit_prompt, build_textual_gradient_prompt
from .schema import (
    BeamSearchState,
    MAX_GRADIENTS,
    MAX_REPORTS,
    MAX_SUGGESTIONS,
    PromptEffectivenessReport,
    PromptOptimizationSuggestion,
    PromptSegment,
    RolloutMessage,
    RolloutResult,
    TextualGradient,
    VersionedPromptTemplate,
    default_apo_config,
)


def _now_ms() -> int:
    return int(time.time() * 1000)


class APOService:
    def __init__(
        self,
        trace_collector: TraceCollector,
        storage: Optional[MemoryStorage] = None,
        clock: Optional[Callable[[], int]] = None,
        uuid_fn: Optional[Callable[[], str]] = None,
        optimizer: Optional[Any] = None,
    ) -> None:
        self._traces = trace_collector
        self._storage = storage if storage is not None else MemoryStorage()
        self._clock = clock or _now_ms
        self._uuid = uuid_fn or new_uuid
        self._optimizer = optimizer  # local gradient/beam engine (GPU) or stub
        self._reports: List[PromptEffectivenessReport] = []
        self._suggestions: List[PromptOptimizationSuggestion] = []
        self._segments: List[PromptSegment] = []
        self._config: Dict[str, Any] = default_apo_config()
        self._beam_state: Optional[BeamSearchState] = None
        self._textual_gradients: List[TextualGradient] = []
        self._dirty = False
        self._suggestion_listeners: List[Callable[[List[PromptOptimizationSuggestion]], None]] = []
        self._load_from_storage()

    # --- events ---

    def on_did_generate_suggestions(self, fn: Callable[[List[PromptOptimizationSuggestion]], None]) -> None:
        self._suggestion_listeners.append(fn)

    def _fire_suggestions(self, suggestions: List[PromptOptimizationSuggestion]) -> None:
        if not suggestions:
            return
        for fn in self._suggestion_listeners:
            try:
                fn(suggestions)
            except Exception:
                pass

    # --- persistence (same keys/shapes as the reference) ---

    def _load_from_storage(self) -> None:
        try:
            cfg_json = self._storage.get(APO_CONFIG_KEY)
            if cfg_json:
                self._config = {**default_apo_config(), **js_parse(cfg_json)}
            data_json = self._storage.get(APO_STORAGE_KEY)
            if data_json:
                data = js_parse(data_json)
                self._reports = []  # reports kept as raw dicts is enough for queries; keep typed ones fresh
                self._raw_reports = data.get("reports") or []
                self._suggestions = [PromptOptimizationSuggestion.from_json(s) for s in (data.get("suggestions") or [])]
            else:
                self._raw_reports = []
            seg_json = self._storage.get(APO_SEGMENTS_KEY)
            if seg_json:
                self._segments = [PromptSegment.from_json(s) for s in js_parse(seg_json)]
            beam_json = self._storage.get(APO_BEAM_STATE_KEY)
            if beam_json:
                self._beam_state = BeamSearchState.from_json(js_parse(beam_json))
            grad_json = self._storage.get(APO_GRADIENTS_KEY)
            if grad_json:
                self._textual_gradients = [TextualGradient.from_json(g) for g in js_parse(grad_json)]
        except Exception:
            self._raw_reports = getattr(self, "_raw_reports", [])

    def flush(self) -> None:
        if not self._dirty:
            return
        try:
            raw_reports = getattr(self, "_raw_reports", []) + [r.to_json() for r in self._reports]
            if len(raw_reports) > MAX_REPORTS:
                raw_reports = raw_reports[-MAX_REPORTS:]
            if len(self._suggestions) > MAX_SUGGESTIONS:
                self._suggestions = self._suggestions[-MAX_SUGGESTIONS:]
            self._storage.store(APO_STORAGE_KEY, js_stringify({
                "reports": raw_reports,
                "suggestions": [s.to_json() for s in self._suggestions],
            }))
            self._storage.store(APO_SEGMENTS_KEY, js_stringify([s.to_json() for s in self._segments]))
            if self._beam_state is not None:
                self._storage.store(APO_BEAM_STATE_KEY, js_stringify(self._beam_state.to_json()))
            if self._textual_gradients:
                if len(self._textual_gradients) > MAX_GRADIENTS:
                    self._textual_gradients = self._textual_gradients[-MAX_GRADIENTS:]
                self._storage.store(APO_GRADIENTS_KEY, js_stringify([g.to_json() for g in self._textual_gradients]))
            self._dirty = False
        except Exception:
            pass
        if hasattr(self._storage, "flush"):
            self._storage.flush()

    # --- auto-analysis gates (reference _tryAutoAnalyze :454-473) ---

    def should_auto_analyze(self) -> bool:
        if not (self._config["enabled"] and self._config["autoAnalyzeEnabled"]):
            return False
        stats = self._traces.get_stats()
        if stats["totalTraces"] < self._config["minTracesForAnalysis"]:
            return False
        if stats["totalFeedbacks"] < self._config["minFeedbacksForAnalysis"]:
            return False
        last = self._reports[-1] if self._reports else None
        if last and (self._clock() - last.generated_at) < self._config["autoAnalyzeIntervalMs"]:
            return False
        return True

    def try_auto_analyze(self) -> Optional[PromptEffectivenessReport]:
        """One auto-analysis tick.  Returns the report if analysis ran.

        When goodRate < 0.7 and feedbacks >= 15, also triggers a textual
        gradient through the local optimizer (reference gate :468).
        """
        if not self.should_auto_analyze():
            return None
        report = self.analyze_prompt_effectiveness()
        stats = self._traces.get_stats()
        if report.good_rate < 0.7 and self._config["uploadOptimizationsToServer"] and stats["totalFeedbacks"] >= 15:
            try:
                self.request_textual_gradient()
            except Exception:
                pass
        return report

    # --- report builder (reference _buildReport :498-625) ---

    def analyze_prompt_effectiveness(self) -> PromptEffectivenessReport:
        traces = self._traces.get_all_traces()
        report = self._build_report(traces)
        self._reports.append(report)
        self._dirty = True
        self.flush()
        return report

    def _extract_mode(self, trace: ConversationTrace) -> str:
        if trace.metadata and trace.metadata.get("chatMode"):
            return str(trace.metadata["chatMode"])
        return "unknown"

    def _build_report(self, traces: List[ConversationTrace]) -> PromptEffectivenessReport:
        now = self._clock()
        good = bad = none = 0
        by_mode: Dict[str, Dict[str, float]] = {}
        oldest = float("inf")
        newest = 0
        for t in traces:
            if t.start_time < oldest:
                oldest = t.start_time
            if t.start_time > newest:
                newest = t.start_time
            fb = t.summary.user_feedback
            if fb == "good":
                good += 1
            elif fb == "bad":
                bad += 1
            else:
                none += 1
            mode = self._extract_mode(t)
            m = by_mode.setdefault(mode, {"total": 0, "good": 0, "bad": 0, "goodRate": 0})
            m["total"] += 1
            if fb == "good":
                m["good"] += 1
            if fb == "bad":
                m["bad"] += 1
        for m in by_mode.values():
            tot = m["good"] + m["bad"]
            m["goodRate"] = m["good"] / tot if tot > 0 else 0

        total_with_fb = good + bad
        good_rate = good / total_with_fb if total_with_fb > 0 else 0

        with_reward = [t for t in traces if t.summary.final_reward is not None]
        avg_reward = (sum(t.summary.final_reward or 0 for t in with_reward) / len(with_reward)) if with_reward else None
        reward_by_dim: Dict[str, Dict[str, float]] = {}
        for t in with_reward:
            for d in t.summary.reward_dimensions:
                a = reward_by_dim.setdefault(d.name, {"sum": 0.0, "count": 0, "avg": 0.0})
                a["sum"] += d.value
                a["count"] += 1
        for a in reward_by_dim.values():
            a["avg"] = a["sum"] / a["count"] if a["count"] > 0 else 0

        patterns = analyze_patterns(traces, self._uuid)
        patterns.extend(reward_dimension_patterns(reward_by_dim, self._uuid))
        suggestions = self._generate_local_suggestions(good_rate, patterns, by_mode, avg_reward, reward_by_dim)

        report = PromptEffectivenessReport(
            id=self._uuid(),
            generated_at=now,
            period={"from": now if oldest == float("inf") else int(oldest), "to": int(newest) or now},
            total_conversations=len(traces),
            good_feedback_count=good,
            bad_feedback_count=bad,
            no_feedback_count=none,
            good_rate=good_rate,
            by_mode=by_mode,
            patterns=patterns,
            suggestions=suggestions,
        )
        self._suggestions.extend(suggestions)
        self._fire_suggestions(suggestions)
        return report

    def _generate_local_suggestions(
        self,
        good_rate: float,
        patterns,
        by_mode: Dict[str, Dict[str, float]],
        avg_reward: Optional[float] = None,
        reward_by_dimension: Optional[Dict[str, Dict[str, float]]] = None,
    ) -> List[PromptOptimizationSuggestion]:
        """Reference _generateLocalSuggestions (apoService.ts:775-862)."""
        suggestions: List[PromptOptimizationSuggestion] = []
        if 0 < good_rate < 0.5:
            reward_info = f" (avg reward: {to_fixed(avg_reward, 3)})" if avg_reward is not None else ""
            suggestions.append(PromptOptimizationSuggestion(
                id=self._uuid(), target_category="core_behavior", type="modify", priority="high",
                description=f"Overall approval rate is only {to_fixed(good_rate * 100, 1)}%{reward_info}, comprehensive prompt optimization needed",
                reasoning="Approval rate below 50% indicates systemic issues with current prompt, recommend requesting backend APO service for deep optimization",
                estimated_impact="Expected to improve approval rate by 10-20%",
            ))
        if reward_by_dimension:
            for dim_name, st in reward_by_dimension.items():
                if st["avg"] < 0 and st["count"] >= 3:
                    target = DIMENSION_CATEGORY_MAP.get(dim_name, "core_behavior")
                    suggestions.append(PromptOptimizationSuggestion(
                        id=self._uuid(), target_category=target, type="modify",
                        priority="high" if st["avg"] < -0.5 else "medium",
                        description=f"{dim_name} dimension performing poorly (avg: {to_fixed(st['avg'], 3)}, n={int(st['count'])})",
                        reasoning=f"This reward dimension is consistently negative, indicating prompt guidance needs improvement for {dim_name}",
                        estimated_impact=f"Expected to improve {dim_name} dimension reward by 0.2-0.5",
                    ))
        for p in patterns:
            if p.severity == "high":
                suggestions.append(PromptOptimizationSuggestion(
                    id=self._uuid(), target_category=p.related_category, type="modify", priority="high",
                    description=f"High-frequency issue: {p.description} (occurred {p.frequency} times)",
                    reasoning="This problem pattern occurs frequently with high severity, targeted optimization of related prompt rules needed",
                    estimated_impact=f"Expected to reduce {min(p.frequency, 5)} similar issues",
                ))
        for mode, st in by_mode.items():
            if st["total"] >= 5 and st["goodRate"] < 0.3:
                suggestions.append(PromptOptimizationSuggestion(
                    id=self._uuid(), target_category="mode_specific", type="modify", priority="medium",
                    description=f"{mode} mode approval rate is only {to_fixed(st['goodRate'] * 100, 1)}%, prompt optimization needed for this mode",
                    reasoning="This mode's approval rate is significantly below average, mode-specific prompt rules may need adjustment",
                    estimated_impact=f"Expected to improve {mode} mode approval rate",
                ))
        return suggestions

    # --- rollout conversion (reference _convertTracesToRolloutResults :866-914) ---

    def convert_traces_to_rollout_results(self, traces: List[ConversationTrace]) -> List[RolloutResult]:
        out: List[RolloutResult] = []
        for t in traces:
            messages: List[RolloutMessage] = []
            for sp in t.spans:
                if sp.type == "user_message":
                    messages.append(RolloutMessage("user", sp.data.get("contentPreview") or ""))
                elif sp.type == "assistant_message":
                    messages.append(RolloutMessage("assistant", sp.data.get("contentPreview") or ""))
                elif sp.type == "tool_call":
                    messages.append(RolloutMessage(
                        "tool", sp.data.get("toolResult") or "",
                        tool_name=sp.data.get("toolName"), tool_success=sp.data.get("toolSuccess"),
                    ))
            fb = t.summary.user_feedback
            status = "succeeded" if fb == "good" else "failed" if fb == "bad" else "failed" if t.summary.has_errors else "unknown"
            sm = t.summary
            total_tools = sm.tool_calls_succeeded + sm.tool_calls_failed
            out.append(RolloutResult(
                trace_id=t.id, thread_id=t.thread_id, status=status,
                final_reward=sm.final_reward,
                reward_dimensions=[d.to_json() for d in sm.reward_dimensions],
                messages=messages, chat_mode=self._extract_mode(t),
                tool_call_stats={
                    "totalCalls": total_tools,
                    "succeeded": sm.tool_calls_succeeded,
                    "failed": sm.tool_calls_failed,
                    "successRate": sm.tool_calls_succeeded / total_tools if total_tools > 0 else None,
                    "byToolName": sm.tool_calls_by_name,
                    "totalDurationMs": sm.total_tool_duration_ms,
                },
                llm_stats={"totalCalls": sm.total_llm_calls, "totalTokens": sm.total_tokens},
            ))
        return out

    def recent_rollouts(self, limit: int) -> List[RolloutResult]:
        traces = [t for t in self._traces.get_all_traces() if t.summary.user_feedback is not None]
        traces.sort(key=lambda t: t.start_time, reverse=True)
        return self.convert_traces_to_rollout_results(traces[:limit])

    # --- textual gradient (local replacement for POST /api/apo/gradient) ---

    def request_textual_gradient(self) -> Optional[TextualGradient]:
        """Run one textual-gradient step through the local optimizer.

        The reference built gradient + apply-edit prompts and POSTed them to
        its backend (apoService.ts:1268-1343); the optimizer here runs the
        same prompts through the local MI355X Llama backbone.
        """
        rollouts = self.recent_rollouts(self._config["gradientBatchSize"])
        if len(rollouts) < 2:
            return None
        current_rules = self.get_optimized_rules()
        gradient_prompt = build_textual_gradient_prompt(current_rules, rollouts)
        apply_edit_template = build_apply_edit_prompt(current_rules, "{{critique_placeholder}}")
        if self._optimizer is None:
            return None
        result = self._optimizer.textual_gradient(
            gradient_prompt=gradient_prompt,
            apply_edit_prompt=apply_edit_template,
            rollout_results=rollouts,
            current_rules=current_rules,
        )
        if not result or not result.get("critique"):
            return None
        avg = sum((r.final_reward or 0) for r in rollouts) / len(rollouts)
        tg = TextualGradient(
            id=self._uuid(),
            prompt_version=(self._beam_state.history_best_prompt.version
                            if self._beam_state and self._beam_state.history_best_prompt else "v0"),
            critique=result["critique"],
            rollout_summary=f"Based on {len(rollouts)} rollouts, avg reward: {to_fixed(avg, 3)}",
            created_at=self._clock(),
        )
        self._textual_gradients.append(tg)
        if result.get("editedPrompt"):
            suggestion = PromptOptimizationSuggestion(
                id=self._uuid(), target_category="core_behavior", type="modify", priority="high",
                description=f"Textual Gradient: {tg.critique[:100]}...",
                suggested_content=result["editedPrompt"],
                reasoning=tg.critique,
                estimated_impact="Prompt optimization based on Textual Gradient",
                prompt_version=tg.prompt_version,
            )
            self._suggestions.append(suggestion)
            self._fire_suggestions([suggestion])
        self._dirty = True
        self.flush()
        return tg

    # --- beam search (local replacement for POST /api/apo/optimize) ---

    def ensure_beam_state(self) -> BeamSearchState:
        if self._beam_state is None:
            self._beam_state = BeamSearchState(
                current_round=0,
                total_rounds=self._config["beamRounds"],
                beam=[],
                history_best_prompt=None,
                history_best_score=float("-inf"),
                version_counter=0,
                started_at=self._clock(),
                last_updated_at=self._clock(),
            )
        return self._beam_state

    def apply_beam_update(self, beam_update: Dict[str, Any]) -> None:
        """Reference beamUpdate handling (apoService.ts:1139-1166)."""
        state = self.ensure_beam_state()
        if beam_update.get("beam") is not None:
            state.beam = [
                b if isinstance(b, VersionedPromptTemplate) else VersionedPromptTemplate.from_json(b)
                for b in beam_update["beam"]
            ]
        if beam_update.get("round") is not None:
            state.current_round = beam_update["round"]
        best = beam_update.get("bestPrompt")
        best_score = beam_update.get("bestScore")
        if best is not None and best_score is not None and best_score > state.history_best_score:
            if not isinstance(best, VersionedPromptTemplate):
                best = VersionedPromptTemplate.from_json(best)
            state.history_best_prompt = best
            state.history_best_score = best_score
            self.apply_beam_best_prompt(best)
        state.last_updated_at = self._clock()
        self._dirty = True

    def apply_beam_best_prompt(self, best_prompt: VersionedPromptTemplate) -> None:
        """Reference _applyBeamBestPrompt (apoService.ts:1219-1264)."""
        rules = [ln for ln in best_prompt.content.split("\n") if ln.strip().startswith("- ")]
        now = self._clock()
        if not rules:
            existing = next((s for s in self._segments if s.category == "core_behavior" and s.is_active), None)
            if existing:
                existing.original_content = existing.original_content or existing.content
                existing.content = best_prompt.content
                existing.is_optimized = True
                existing.version += 1
                existing.updated_at = now
            else:
                self._segments.append(PromptSegment(
                    id=self._uuid(), category="core_behavior", content=best_prompt.content,
                    is_active=True, is_optimized=True, version=1, created_at=now, updated_at=now,
                ))
        else:
            for rule in rules:
                # JS: rule.replace(/^-\s*/, '').trim()
                import re
                content = re.sub(r"^-\s*", "", rule).strip()
                if not content:
                    continue
                if not any(s.is_active and s.content == content for s in self._segments):
                    self._segments.append(PromptSegment(
                        id=self._uuid(), category="core_behavior", content=content,
                        is_active=True, is_optimized=True, version=1, created_at=now, updated_at=now,
                    ))
        self._dirty = True

    def record_textual_gradient(self, critique: str, rollout_count: int) -> TextualGradient:
        tg = TextualGradient(
            id=self._uuid(),
            prompt_version=(self._beam_state.history_best_prompt.version
                            if self._beam_state and self._beam_state.history_best_prompt else "v0"),
            critique=critique,
            rollout_summary=f"Based on {rollout_count} rollouts",
            created_at=self._clock(),
        )
        self._textual_gradients.append(tg)
        if len(self._textual_gradients) > MAX_GRADIENTS:
            self._textual_gradients = self._textual_gradients[-MAX_GRADIENTS:]
        self._dirty = True
        return tg

    # --- segment management (reference :1358-1458) ---

    def get_active_segments(self) -> List[PromptSegment]:
        return [s for s in self._segments if s.is_active]

    def get_optimized_prompt_for_category(self, category: str) -> Optional[str]:
        for s in self._segments:
            if s.is_active and s.is_optimized and s.category == category:
                return s.content
        return None

    def get_optimized_rules(self) -> List[str]:
        return [s.content for s in self._segments if s.is_active and s.is_optimized]

    def add_segment(self, category: str, content: str, is_optimized: bool = False) -> PromptSegment:
        now = self._clock()
        seg = PromptSegment(
            id=self._uuid(), category=category, content=content, is_active=True,
            is_optimized=is_optimized, version=1, created_at=now, updated_at=now,
        )
        self._segments.append(seg)
        self._dirty = True
        return seg

    def apply_suggestion(self, suggestion_id: str) -> None:
        s = next((x for x in self._suggestions if x.id == suggestion_id), None)
        if not s or s.status != "pending":
            return
        s.status = "applied"
        s.applied_at = self._clock()
        if s.suggested_content:
            if s.target_segment_id:
                existing = next((x for x in self._segments if x.id == s.target_segment_id), None)
            else:
                existing = next((x for x in self._segments if x.category == s.target_category and x.is_active), None)
            if existing and s.type == "modify":
                existing.original_content = existing.original_content or existing.content
                existing.content = s.suggested_content
                existing.is_optimized = True
                existing.version += 1
                existing.updated_at = self._clock()
            elif s.type == "add":
                now = self._clock()
                self._segments.append(PromptSegment(
                    id=self._uuid(), category=s.target_category, content=s.suggested_content,
                    is_active=True, is_optimized=True, version=1, created_at=now, updated_at=now,
                ))
        self._dirty = True
        self.flush()

    def reject_suggestion(self, suggestion_id: str) -> None:
        s = next((x for x in self._suggestions if x.id == suggestion_id), None)
        if not s or s.status != "pending":
            return
        s.status = "rejected"
        self._dirty = True
        self.flush()

    def revert_suggestion(self, suggestion_id: str) -> None:
        s = next((x for x in self._suggestions if x.id == suggestion_id), None)
        if not s or s.status != "applied":
            return
        if s.target_segment_id:
            seg = next((x for x in self._segments if x.id == s.target_segment_id), None)
            if seg and seg.original_content:
                seg.content = seg.original_content
                seg.original_content = None
                seg.is_optimized = False
                seg.version += 1
                seg.updated_at = self._clock()
        elif s.type == "modify":
            seg = next((x for x in self._segments if x.category == s.target_category and x.is_active and x.is_optimized), None)
            if seg and seg.original_content:
                seg.content = seg.original_content
                seg.original_content = None
                seg.is_optimized = False
                seg.version += 1
                seg.updated_at = self._clock()
        elif s.type == "add":
            self._segments = [
                x for x in self._segments
                if not (x.category == s.target_category and x.is_optimized and x.content == s.suggested_content)
            ]
        s.status = "reverted"
        self._dirty = True
        self.flush()

    # --- queries ---

    def get_latest_report(self) -> Optional[PromptEffectivenessReport]:
        return self._reports[-1] if self._reports else None

    def get_pending_suggestions(self) -> List[PromptOptimizationSuggestion]:
        return [s for s in self._suggestions if s.status == "pending"]

    def get_beam_state(self) -> Optional[BeamSearchState]:
        return self._beam_state

    def get_textual_gradients(self, limit: Optional[int] = None) -> List[TextualGradient]:
        gradients = sorted(self._textual_gradients, key=lambda g: g.created_at, reverse=True)
        return gradients[:limit] if limit else gradients

    def get_stats(self) -> Dict[str, Any]:
        """Reference getStats (apoService.ts:1470-1516) — same keys."""
        applied = sum(1 for s in self._suggestions if s.status == "applied")
        rejected = sum(1 for s in self._suggestions if s.status == "rejected")
        active = sum(1 for s in self._segments if s.is_active)
        optimized = sum(1 for s in self._segments if s.is_active and s.is_optimized)
        latest = self._reports[-1] if self._reports else None
        avg_final: Optional[float] = None
        try:
            recent = [t for t in self._traces.get_all_traces() if t.summary.final_reward is not None]
            recent.sort(key=lambda t: t.start_time, reverse=True)
            recent = recent[:20]
            if recent:
                avg_final = sum(t.summary.final_reward or 0 for t in recent) / len(recent)
        except Exception:
            pass
        best = self._beam_state.history_best_score if self._beam_state else None
        return {
            "totalReports": len(self._reports) + len(getattr(self, "_raw_reports", [])),
            "totalSuggestions": len(self._suggestions),
            "appliedSuggestions": applied,
            "rejectedSuggestions": rejected,
            "activeSegments": active,
            "optimizedSegments": optimized,
            "lastAnalysisTime": latest.generated_at if latest else None,
            "currentGoodRate": latest.good_rate if latest else None,
            "beamSearchActive": self._beam_state is not None,
            "beamCurrentRound": self._beam_state.current_round if self._beam_state else None,
            "beamBestScore": best if (best is not None and best != float("-inf")) else None,
            "totalTextualGradients": len(self._textual_gradients),
            "avgFinalReward": avg_final,
        }

    def get_config(self) -> Dict[str, Any]:
        return dict(self._config)

    def set_config(self, config: Dict[str, Any]) -> None:
        self._config = {**self._config, **config}
        self._storage.store(APO_CONFIG_KEY, js_stringify(self._config))
