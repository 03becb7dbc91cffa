"""TraceCollector — conversation-turn span collection + reward aggregation.

Capability- and format-compatible rebuild of the reference's TraceCollectorService
(reference: common/traceCollectorService.ts).  Differences by design:

- The reference records spans via ``queueMicrotask`` on a single JS event loop so
  the agent loop never blocks; here every ``record_*`` appends to a lock-free-ish
  in-process queue that is drained inline (O(µs)) and persistence happens on a
  background flush (30 s interval equivalent, driven by ``maybe_flush``/``flush``).
- ``uploadToServer`` posted to ``{apiBaseUrl}/api/traces`` over HTTPS
  (traceCollectorService.ts:797-899); here ``build_upload_payload`` produces the
  identical version-'2.0.0' payload and the transport is pluggable (file / HTTP
  sink), keeping the incremental ``uploadedIds`` watermark semantics.

JSON shapes (spans, traces, export envelope v1.0.0, upload payload v2.0.0) and
storage keys are byte-compatible with the reference.
"""

from __future__ import annotations

import datetime
import threading
import time
from typing import Any, Callable, Dict, List, Optional

from ..storage import (
    MemoryStorage,
    TRACE_FEEDBACK_KEY,
    TRACE_STORAGE_KEY,
    TRACE_UPLOADED_IDS_KEY,
    TRACE_UPLOAD_CONFIG_KEY,
)
from ..utils.jsonutil import js_parse, js_stringify
from .reward import compute_reward_signals
from .schema import (
    ConversationTrace,
    FLUSH_INTERVAL_MS,
    MAX_SPANS_PER_TRACE,
    MAX_TRACES,
    TraceSpan,
    TraceSummary,
    new_uuid,
    truncate,
)


def _now_ms() -> int:
    return int(time.time() * 1000)


class TraceCollector:
    """Span collection per conversation turn; 9-dim reward; stats; persistence."""

    def __init__(
        self,
        storage: Optional[MemoryStorage] = None,
        clock: Optional[Callable[[], int]] = None,
        uuid_fn: Optional[Callable[[], str]] = None,
    ) -> None:
        self._storage = storage if storage is not None else MemoryStorage()
        self._clock = clock or _now_ms
        self._uuid = uuid_fn or new_uuid
        self._traces: Dict[str, ConversationTrace] = {}
        self._active_traces: Dict[str, str] = {}  # thread_id -> trace_id
        self._feedbacks: Dict[str, Optional[str]] = {}  # "threadId:msgIdx" -> feedback
        self._dirty = False
        self._lock = threading.RLock()
        self._last_flush_ms = self._clock()
        self._uploaded_ids: set = set()
        self._auto_upload = {"enabled": False, "intervalMs": 300000}
        self._state_listeners: List[Callable[[], None]] = []
        self._load_from_storage()
        self._load_upload_config()
        self._load_uploaded_ids()

    # --- events ---

    def on_did_change_state(self, fn: Callable[[], None]) -> None:
        self._state_listeners.append(fn)

    def _fire_state_change(self) -> None:
        for fn in self._state_listeners:
            try:
                fn()
            except Exception:
                pass

    # --- internal ---

    def _feedback_key(self, thread_id: str, message_idx: int) -> str:
        return f"{thread_id}:{message_idx}"

    def _get_or_create_trace(self, thread_id: str) -> ConversationTrace:
        trace_id = self._active_traces.get(thread_id)
        if trace_id and trace_id in self._traces:
            return self._traces[trace_id]
        new_id = self.start_trace(thread_id)
        return self._traces[new_id]

    def _add_span(self, trace: ConversationTrace, span: TraceSpan) -> None:
        if len(trace.spans) >= MAX_SPANS_PER_TRACE:
            return
        trace.spans.append(span)
        self._dirty = True

    def _create_span(self, trace_id: str, thread_id: str, message_idx: int, type_: str, data: Dict[str, Any]) -> TraceSpan:
        return TraceSpan(
            id=self._uuid(),
            trace_id=trace_id,
            thread_id=thread_id,
            message_idx=message_idx,
            type=type_,
            timestamp=self._clock(),
            data=data,
        )

    # --- storage ---

    def _load_from_storage(self) -> None:
        try:
            traces = js_parse(self._storage.get(TRACE_STORAGE_KEY, "[]") or "[]")
            for t in traces:
                trace = ConversationTrace.from_json(t)
                self._traces[trace.id] = trace
            feedbacks = js_parse(self._storage.get(TRACE_FEEDBACK_KEY, "{}") or "{}")
            for k, v in feedbacks.items():
                self._feedbacks[k] = v
        except Exception:
            pass  # silent, as in the reference

    def _load_upload_config(self) -> None:
        try:
            cfg_json = self._storage.get(TRACE_UPLOAD_CONFIG_KEY)
            if cfg_json:
                cfg = js_parse(cfg_json)
                self._auto_upload = {
                    "enabled": bool(cfg.get("enabled", False)),
                    "intervalMs": cfg.get("intervalMs", 300000) or 300000,
                }
        except Exception:
            pass

    def _load_uploaded_ids(self) -> None:
        try:
            ids_json = self._storage.get(TRACE_UPLOADED_IDS_KEY)
            if ids_json:
                self._uploaded_ids = set(js_parse(ids_json))
        except Exception:
            pass

    def flush(self) -> None:
        """Persist (the reference's _saveToStorage, run every 30 s + on dispose)."""
        with self._lock:
            if not self._dirty:
                return
            all_traces = list(self._traces.values())
            if len(all_traces) > MAX_TRACES:
                all_traces.sort(key=lambda t: t.start_time or 0, reverse=True)
                keep = all_traces[:MAX_TRACES]
                self._traces = {t.id: t for t in keep}
                all_traces = keep
            self._storage.store(TRACE_STORAGE_KEY, js_stringify([t.to_json() for t in self._traces.values()]))
            self._storage.store(TRACE_FEEDBACK_KEY, js_stringify(self._feedbacks))
            self._dirty = False
            self._last_flush_ms = self._clock()
        if hasattr(self._storage, "flush"):
            self._storage.flush()

    def maybe_flush(self) -> None:
        if self._clock() - self._last_flush_ms >= FLUSH_INTERVAL_MS:
            self.flush()

    # --- trace lifecycle ---

    def start_trace(self, thread_id: str, metadata: Optional[Dict[str, Any]] = None) -> str:
        with self._lock:
            trace_id = self._uuid()
            trace = ConversationTrace(
                id=trace_id,
                thread_id=thread_id,
                start_time=self._clock(),
                spans=[],
                metadata=metadata,
                summary=TraceSummary(),
            )
            self._traces[trace_id] = trace
            self._active_traces[thread_id] = trace_id
            self._dirty = True
            return trace_id

    def end_trace(self, trace_id: str) -> None:
        with self._lock:
            trace = self._traces.get(trace_id)
            if trace:
                trace.end_time = self._clock()
                compute_reward_signals(trace)
                self._dirty = True
        self.flush()

    def end_trace_for_thread(self, thread_id: str) -> None:
        trace_id = self._active_traces.get(thread_id)
        if trace_id:
            self.end_trace(trace_id)

    # --- span recording (fire-and-forget; errors swallowed like the reference) ---

    def record_user_message(self, thread_id: str, message_idx: int, content: str) -> None:
        try:
            with self._lock:
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "user_message", {
                    "contentPreview": truncate(content),
                    "contentLength": len(content),
                })
                self._add_span(trace, span)
        except Exception:
            pass

    def record_assistant_message(self, thread_id: str, message_idx: int, content: str,
                                 model: Optional[str] = None, provider: Optional[str] = None) -> None:
        try:
            with self._lock:
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "assistant_message", {
                    "contentPreview": truncate(content),
                    "contentLength": len(content),
                    "model": model,
                    "provider": provider,
                })
                self._add_span(trace, span)
        except Exception:
            pass

    def record_llm_call(self, thread_id: str, message_idx: int, *, model: Optional[str] = None,
                        provider: Optional[str] = None, input_tokens: Optional[int] = None,
                        output_tokens: Optional[int] = None, temperature: Optional[float] = None,
                        duration: Optional[float] = None) -> None:
        try:
            with self._lock:
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "llm_call", {
                    "model": model,
                    "provider": provider,
                    "inputTokens": input_tokens,
                    "outputTokens": output_tokens,
                    "temperature": temperature,
                })
                span.duration = duration
                self._add_span(trace, span)
                trace.summary.total_llm_calls += 1
                trace.summary.total_tokens += (input_tokens or 0) + (output_tokens or 0)
        except Exception:
            pass

    def record_tool_call(self, thread_id: str, message_idx: int, *, tool_name: str,
                         tool_params: Optional[str] = None, tool_result: Optional[str] = None,
                         tool_success: bool, duration: Optional[float] = None) -> None:
        try:
            with self._lock:
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "tool_call", {
                    "toolName": tool_name,
                    "toolParams": truncate(tool_params),
                    "toolResult": truncate(tool_result),
                    "toolSuccess": tool_success,
                })
                span.duration = duration
                self._add_span(trace, span)
                s = trace.summary
                s.total_tool_calls += 1
                if tool_success:
                    s.tool_calls_succeeded += 1
                else:
                    s.tool_calls_failed += 1
                stats = s.tool_calls_by_name.get(tool_name) or {"total": 0, "succeeded": 0, "failed": 0}
                stats["total"] += 1
                if tool_success:
                    stats["succeeded"] += 1
                else:
                    stats["failed"] += 1
                s.tool_calls_by_name[tool_name] = stats
                if duration and duration > 0:
                    s.total_tool_duration_ms += duration
                self._dirty = True
        except Exception:
            pass

    def record_user_feedback(self, thread_id: str, message_idx: int, feedback: Optional[str]) -> None:
        try:
            with self._lock:
                key = self._feedback_key(thread_id, message_idx)
                self._feedbacks[key] = feedback
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "user_feedback", {
                    "feedback": feedback,
                })
                self._add_span(trace, span)
                trace.summary.user_feedback = feedback
                self._dirty = True
                compute_reward_signals(trace)
            self._fire_state_change()
            self.flush()
        except Exception:
            pass

    def record_error(self, thread_id: str, message_idx: int, error_message: str) -> None:
        try:
            with self._lock:
                trace = self._get_or_create_trace(thread_id)
                span = self._create_span(trace.id, thread_id, message_idx, "error", {
                    "errorMessage": truncate(error_message, 1000),
                })
                self._add_span(trace, span)
                trace.summary.has_errors = True
        except Exception:
            pass

    # --- queries ---

    def get_feedback(self, thread_id: str, message_idx: int) -> Optional[str]:
        return self._feedbacks.get(self._feedback_key(thread_id, message_idx))

    def get_all_traces(self) -> List[ConversationTrace]:
        return list(self._traces.values())

    def get_stats(self) -> Dict[str, Any]:
        """Reference getStats (traceCollectorService.ts:577-628) — same keys."""
        total_spans = 0
        oldest: Optional[int] = None
        newest: Optional[int] = None
        for t in self._traces.values():
            total_spans += len(t.spans)
            if oldest is None or t.start_time < oldest:
                oldest = t.start_time
            if newest is None or t.start_time > newest:
                newest = t.start_time
        good = sum(1 for v in self._feedbacks.values() if v == "good")
        bad = sum(1 for v in self._feedbacks.values() if v == "bad")
        total_tool = total_ok = total_fail = 0
        reward_sum = 0.0
        with_reward = 0
        for t in self._traces.values():
            total_tool += t.summary.total_tool_calls
            total_ok += t.summary.tool_calls_succeeded
            total_fail += t.summary.tool_calls_failed
            if t.summary.final_reward is not None:
                reward_sum += t.summary.final_reward
                with_reward += 1
        return {
            "totalTraces": len(self._traces),
            "totalSpans": total_spans,
            "totalFeedbacks": good + bad,
            "goodFeedbacks": good,
            "badFeedbacks": bad,
            "storageUsedBytes": self._estimate_storage_bytes(),
            "oldestTraceTime": oldest,
            "newestTraceTime": newest,
            "totalToolCalls": total_tool,
            "totalToolSucceeded": total_ok,
            "totalToolFailed": total_fail,
            "toolSuccessRate": (total_ok / total_tool) if total_tool > 0 else None,
            "avgFinalReward": (reward_sum / with_reward) if with_reward > 0 else None,
            "tracesWithReward": with_reward,
        }

    def _estimate_storage_bytes(self) -> int:
        try:
            traces_json = js_stringify([t.to_json() for t in self._traces.values()])
            feedbacks_json = js_stringify(self._feedbacks)
            return len(traces_json) + len(feedbacks_json)
        except Exception:
            return 0

    def export_data(self) -> str:
        """Versioned '1.0.0' export envelope (reference exportData :634-642)."""
        return js_stringify({
            "version": "1.0.0",
            "exportTime": datetime.datetime.fromtimestamp(
                self._clock() / 1000, tz=datetime.timezone.utc
            ).isoformat(timespec="milliseconds").replace("+00:00", "Z"),
            "stats": self.get_stats(),
            "traces": [t.to_json() for t in self._traces.values()],
            "feedbacks": self._feedbacks,
        }, indent=2)

    def import_data(self, text: str) -> int:
        """Load a reference (or our) export envelope; returns traces imported."""
        data = js_parse(text)
        traces = data.get("traces", data if isinstance(data, list) else [])
        n = 0
        with self._lock:
            for t in traces:
                trace = ConversationTrace.from_json(t)
                self._traces[trace.id] = trace
                n += 1
            for k, v in (data.get("feedbacks", {}) or {}).items():
                self._feedbacks[k] = v
            self._dirty = True
        return n

    def clear_all_data(self) -> None:
        with self._lock:
            self._traces.clear()
            self._active_traces.clear()
            self._feedbacks.clear()
            self._dirty = True
        self.flush()
        self._fire_state_change()

    # --- upload (payload-compatible; transport pluggable) ---

    def build_upload_payload(self) -> Optional[Dict[str, Any]]:
        """Version-'2.0.0' incremental upload payload (reference :797-899)."""
        new_traces = [t for t in self._traces.values() if t.id not in self._uploaded_ids]
        if not new_traces:
            return None
        with_reward = [t for t in new_traces if t.summary.final_reward is not None]
        avg_final = (sum(t.summary.final_reward or 0 for t in with_reward) / len(with_reward)) if with_reward else None
        total_ok = total_fail = 0
        total_dur = 0.0
        by_name: Dict[str, Dict[str, int]] = {}
        for t in new_traces:
            total_ok += t.summary.tool_calls_succeeded
            total_fail += t.summary.tool_calls_failed
            total_dur += t.summary.total_tool_duration_ms
            for name, st in t.summary.tool_calls_by_name.items():
                agg = by_name.setdefault(name, {"total": 0, "succeeded": 0, "failed": 0})
                agg["total"] += st["total"]
                agg["succeeded"] += st["succeeded"]
                agg["failed"] += st["failed"]
        dim_agg: Dict[str, Dict[str, float]] = {}
        for t in with_reward:
            for d in t.summary.reward_dimensions:
                a = dim_agg.setdefault(d.name, {"sum": 0.0, "count": 0})
                a["sum"] += d.value
                a["count"] += 1
        dim_avg = {name: (a["sum"] / a["count"] if a["count"] > 0 else 0) for name, a in dim_agg.items()}
        thread_ids = {t.thread_id for t in new_traces}
        return {
            "version": "2.0.0",
            "uploadTime": datetime.datetime.fromtimestamp(
                self._clock() / 1000, tz=datetime.timezone.utc
            ).isoformat(timespec="milliseconds").replace("+00:00", "Z"),
            "traces": [t.to_json() for t in new_traces],
            "feedbacks": {k: v for k, v in self._feedbacks.items() if k.split(":")[0] in thread_ids},
            "rewardSummary": {
                "totalTracesWithReward": len(with_reward),
                "avgFinalReward": avg_final,
                "rewardDimensionAvg": dim_avg,
            },
            "toolCallSummary": {
                "totalToolCalls": total_ok + total_fail,
                "totalSucceeded": total_ok,
                "totalFailed": total_fail,
                "successRate": (total_ok / (total_ok + total_fail)) if (total_ok + total_fail) > 0 else None,
                "totalDurationMs": total_dur,
                "byToolName": by_name,
            },
        }

    def upload_to_sink(self, sink: Callable[[Dict[str, Any]], bool]) -> Dict[str, Any]:
        """Run one incremental upload through ``sink`` (returns reference-shaped result)."""
        payload = self.build_upload_payload()
        if payload is None:
            return {"success": True, "message": "No new traces to upload", "uploadedCount": 0}
        try:
            ok = sink(payload)
        except Exception as e:  # transport failure
            return {"success": False, "message": f"Upload failed: {e}", "uploadedCount": 0}
        if not ok:
            return {"success": False, "message": "Sink rejected payload", "uploadedCount": 0}
        for t in payload["traces"]:
            self._uploaded_ids.add(t["id"])
        valid = [i for i in self._uploaded_ids if i in self._traces]
        self._uploaded_ids = set(valid)
        self._storage.store(TRACE_UPLOADED_IDS_KEY, js_stringify(valid))
        return {"success": True, "message": "Upload successful", "uploadedCount": len(payload["traces"])}

    def set_auto_upload_config(self, enabled: bool, interval_ms: Optional[int] = None) -> None:
        self._auto_upload = {"enabled": enabled, "intervalMs": interval_ms if interval_ms is not None else 300000}
        self._storage.store(TRACE_UPLOAD_CONFIG_KEY, js_stringify(self._auto_upload))

    def get_auto_upload_config(self) -> Dict[str, Any]:
        return dict(self._auto_upload)
