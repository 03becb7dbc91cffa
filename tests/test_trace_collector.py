"""TraceCollector behavior + persistence-format tests."""

import json

from senweaver_amd.storage import (
    MemoryStorage,
    TRACE_FEEDBACK_KEY,
    TRACE_STORAGE_KEY,
)
from senweaver_amd.trace import TraceCollector, MAX_SPANS_PER_TRACE
from senweaver_amd.utils.jsonutil import js_stringify


def build_collector(fixed_clock, seq_uuid, storage=None):
    return TraceCollector(storage=storage or MemoryStorage(), clock=fixed_clock, uuid_fn=seq_uuid)


def test_span_truncation(fixed_clock, seq_uuid):
    c = build_collector(fixed_clock, seq_uuid)
    c.start_trace("th", {"chatMode": "normal"})
    long_msg = "x" * 600
    c.record_user_message("th", 0, long_msg)
    span = c.get_all_traces()[0].spans[0]
    assert span.data["contentPreview"] == "x" * 500 + "..."
    assert span.data["contentLength"] == 600


def test_span_cap(fixed_clock, seq_uuid):
    c = build_collector(fixed_clock, seq_uuid)
    c.start_trace("th")
    for i in range(MAX_SPANS_PER_TRACE + 50):
        c.record_user_message("th", i, f"m{i}")
    assert len(c.get_all_traces()[0].spans) == MAX_SPANS_PER_TRACE


def test_auto_create_trace_on_record(fixed_clock, seq_uuid):
    c = build_collector(fixed_clock, seq_uuid)
    c.record_user_message("orphan-thread", 0, "hello")
    traces = c.get_all_traces()
    assert len(traces) == 1
    assert traces[0].thread_id == "orphan-thread"


def test_persistence_roundtrip(fixed_clock, seq_uuid):
    storage = MemoryStorage()
    c = build_collector(fixed_clock, seq_uuid, storage)
    tid = c.start_trace("th", {"chatMode": "agent"})
    c.record_user_message("th", 0, "question")
    c.record_tool_call("th", 1, tool_name="run_command", tool_success=False,
                       tool_params='{"cmd":"ls"}', tool_result="err", duration=50)
    c.end_trace(tid)
    c.record_user_feedback("th", 1, "bad")
    c.flush()

    # Reload from the same storage
    c2 = TraceCollector(storage=storage, clock=fixed_clock, uuid_fn=seq_uuid)
    traces = c2.get_all_traces()
    assert len(traces) == 1
    t = traces[0]
    assert t.metadata == {"chatMode": "agent"}
    assert t.summary.user_feedback == "bad"
    assert t.summary.tool_calls_failed == 1
    assert t.summary.tool_calls_by_name == {"run_command": {"total": 1, "succeeded": 1 - 1, "failed": 1}}
    assert t.summary.final_reward is not None
    assert c2.get_feedback("th", 1) == "bad"


def test_stored_json_shape_camelcase(fixed_clock, seq_uuid):
    storage = MemoryStorage()
    c = build_collector(fixed_clock, seq_uuid, storage)
    tid = c.start_trace("th")
    c.record_llm_call("th", 0, model="llama-3-8b", provider="local", input_tokens=10,
                      output_tokens=5, duration=100)
    c.end_trace(tid)
    c.flush()
    raw = json.loads(storage.get(TRACE_STORAGE_KEY))
    assert isinstance(raw, list)
    t = raw[0]
    for key in ("id", "threadId", "startTime", "endTime", "spans", "summary"):
        assert key in t
    span = [s for s in t["spans"] if s["type"] == "llm_call"][0]
    assert span["data"]["inputTokens"] == 10
    assert span["duration"] == 100
    assert "temperature" not in span["data"]  # undefined fields omitted
    summary = t["summary"]
    for key in ("totalLLMCalls", "totalToolCalls", "totalTokens", "userFeedback", "hasErrors",
                "toolCallsSucceeded", "toolCallsFailed", "toolCallsByName", "totalToolDurationMs",
                "finalReward", "rewardDimensions"):
        assert key in summary
    assert summary["totalLLMCalls"] == 1
    assert summary["totalTokens"] == 15


def test_feedback_storage_format(fixed_clock, seq_uuid):
    storage = MemoryStorage()
    c = build_collector(fixed_clock, seq_uuid, storage)
    c.start_trace("threadA")
    c.record_user_feedback("threadA", 3, "good")
    c.flush()
    raw = json.loads(storage.get(TRACE_FEEDBACK_KEY))
    assert raw == {"threadA:3": "good"}


def test_export_envelope(fixed_clock, seq_uuid):
    c = build_collector(fixed_clock, seq_uuid)
    tid = c.start_trace("th")
    c.end_trace(tid)
    data = json.loads(c.export_data())
    assert data["version"] == "1.0.0"
    assert "exportTime" in data and data["exportTime"].endswith("Z")
    assert "stats" in data and "traces" in data and "feedbacks" in data


def test_import_reference_style_export(fixed_clock, seq_uuid):
    # An envelope shaped exactly like the reference's exportData output
    envelope = {
        "version": "1.0.0",
        "exportTime": "2026-01-01T00:00:00.000Z",
        "stats": {},
        "traces": [{
            "id": "ref-trace-1", "threadId": "th-9", "startTime": 1700000000000,
            "endTime": 1700000005000,
            "spans": [{
                "id": "s1", "traceId": "ref-trace-1", "threadId": "th-9", "messageIdx": 0,
                "type": "user_message", "timestamp": 1700000000100,
                "data": {"contentPreview": "hi", "contentLength": 2},
            }],
            "metadata": {"chatMode": "agent"},
            "summary": {
                "totalLLMCalls": 1, "totalToolCalls": 0, "totalTokens": 42,
                "userFeedback": "good", "hasErrors": False,
                "toolCallsSucceeded": 0, "toolCallsFailed": 0, "toolCallsByName": {},
                "totalToolDurationMs": 0, "finalReward": 0.7,
                "rewardDimensions": [{"name": "user_feedback", "value": 1}],
            },
        }],
        "feedbacks": {"th-9:0": "good"},
    }
    c = build_collector(fixed_clock, seq_uuid)
    n = c.import_data(js_stringify(envelope))
    assert n == 1
    t = c.get_all_traces()[0]
    assert t.summary.final_reward == 0.7
    assert c.get_feedback("th-9", 0) == "good"


def test_upload_payload_incremental(fixed_clock, seq_uuid):
    c = build_collector(fixed_clock, seq_uuid)
    tid = c.start_trace("th")
    c.record_user_feedback("th", 0, "good")
    c.end_trace(tid)
    payload = c.build_upload_payload()
    assert payload["version"] == "2.0.0"
    assert len(payload["traces"]) == 1
    assert payload["rewardSummary"]["totalTracesWithReward"] == 1
    assert payload["feedbacks"] == {"th:0": "good"}
    # After upload through a sink, nothing new remains
    result = c.upload_to_sink(lambda p: True)
    assert result["success"] and result["uploadedCount"] == 1
    assert c.build_upload_payload() is None
    assert c.upload_to_sink(lambda p: True)["uploadedCount"] == 0


def test_trace_cap_keeps_newest(seq_uuid):
    state = {"t": 0}

    def clock():
        state["t"] += 1
        return state["t"]

    c = TraceCollector(storage=MemoryStorage(), clock=clock, uuid_fn=seq_uuid)
    from senweaver_amd.trace.schema import MAX_TRACES
    for i in range(MAX_TRACES + 10):
        tid = c.start_trace(f"th{i}")
        c.end_trace(tid)
    c.flush()
    assert len(c.get_all_traces()) == MAX_TRACES
    starts = [t.start_time for t in c.get_all_traces()]
    assert min(starts) > 10  # the oldest 10 were evicted


def test_span_caps_property():
    """Property: arbitrary recorded content respects the reference caps —
    500-char previews, 200 spans per trace, 1000 traces total."""
    from hypothesis import given, settings, strategies as st
    from senweaver_amd.trace import TraceCollector

    @settings(max_examples=20, deadline=None, derandomize=True)
    @given(st.lists(st.text(alphabet="ab", min_size=0, max_size=1200), min_size=1, max_size=30))
    def prop(contents):
        tc = TraceCollector()
        tid = tc.start_trace("t", {})
        for i, content in enumerate(contents):
            tc.record_user_message("t", i, content)
        tc.end_trace(tid)
        trace = tc.get_all_traces()[0]
        assert len(trace.spans) <= 200
        for s in trace.spans:
            preview = s.data.get("contentPreview")
            if preview is not None:
                assert len(preview) <= 500

    prop()
