"""Golden tests for the 9-dimension reward — values hand-computed from the
reference semantics (common/traceCollectorService.ts:668-788)."""

import pytest

from senweaver_amd.trace import TraceCollector, compute_reward_signals
from senweaver_amd.trace.schema import ConversationTrace, TraceSummary


def make_trace(**kw):
    meta = kw.pop("metadata", None)
    t = ConversationTrace(id="t1", thread_id="th1", start_time=1000, metadata=meta)
    t.end_time = kw.pop("end_time", 2000)
    s = t.summary
    for k, v in kw.pop("summary", {}).items():
        setattr(s, k, v)
    return t


def add_msgs(trace, user=1, assistant=1):
    from senweaver_amd.trace.schema import TraceSpan
    for i in range(user):
        trace.spans.append(TraceSpan(id=f"u{i}", trace_id=trace.id, thread_id=trace.thread_id,
                                     message_idx=i, type="user_message", timestamp=0))
    for i in range(assistant):
        trace.spans.append(TraceSpan(id=f"a{i}", trace_id=trace.id, thread_id=trace.thread_id,
                                     message_idx=i, type="assistant_message", timestamp=0))


def test_good_feedback_normal_mode_full_dims():
    t = make_trace(summary=dict(
        user_feedback="good", total_tool_calls=2, tool_calls_succeeded=1, tool_calls_failed=1,
        total_tool_duration_ms=1000.0, total_llm_calls=2, total_tokens=3000,
    ))
    add_msgs(t, 1, 1)
    final, dims = compute_reward_signals(t)
    by_name = {d.name: d.value for d in dims}
    assert by_name["user_feedback"] == 1.0
    assert by_name["task_completion"] == 1.0
    assert by_name["tool_success_rate"] == 0.0       # 0.5*2-1
    assert by_name["tool_call_reliability"] == -0.2  # 1 failed >= minor(1), normal mode
    assert by_name["tool_call_efficiency"] == 1.0    # 2 <= excellent(3)
    assert by_name["tool_duration_efficiency"] == 1.0  # avg 500ms < 1s
    assert by_name["response_efficiency"] == pytest.approx(0.6)  # 1-0.4*(2-1)
    assert by_name["token_efficiency"] == 0.5        # 2000 < 3000 <= 5000
    assert by_name["conversation_efficiency"] == 1.0  # 1 turn <= 2
    # weighted: .25*1+.18*1+.12*0+.08*-0.2+.05*1+.05*1+.08*.6+.08*.5+.11*1 over weight 1.0
    expected = (0.25 * 1 + 0.18 * 1 + 0.12 * 0 + 0.08 * -0.2 + 0.05 * 1 + 0.05 * 1
                + 0.08 * 0.6 + 0.08 * 0.5 + 0.11 * 1) / 1.0
    assert final == pytest.approx(expected)
    assert final == pytest.approx(0.712)


def test_bad_feedback_with_errors_minimal_dims():
    # No tools, no llm calls, no tokens, no messages -> only 2 dims
    t = make_trace(summary=dict(user_feedback="bad", has_errors=True))
    final, dims = compute_reward_signals(t)
    names = [d.name for d in dims]
    assert names == ["user_feedback", "task_completion"]
    # (-1*.25 + -0.5*.18) / (.25+.18)
    assert final == pytest.approx((-0.25 - 0.09) / 0.43)


def test_no_feedback_completed_clean():
    t = make_trace(summary=dict())
    final, dims = compute_reward_signals(t)
    by_name = {d.name: d.value for d in dims}
    assert by_name["user_feedback"] == 0.0
    assert by_name["task_completion"] == 0.8  # ended, no errors
    assert final == pytest.approx((0.0 * 0.25 + 0.8 * 0.18) / 0.43)


def test_unended_trace_neutral_completion():
    t = make_trace(end_time=None, summary=dict())
    final, dims = compute_reward_signals(t)
    by_name = {d.name: d.value for d in dims}
    assert by_name["task_completion"] == 0.5


def test_agent_mode_adaptive_thresholds():
    # Agent mode: 4 failed tools is below severe(5), at moderate(3) -> -0.5;
    # normal mode the same trace would be -1.0 (>= severe 3)
    common = dict(
        user_feedback=None, total_tool_calls=10, tool_calls_succeeded=6, tool_calls_failed=4,
        total_llm_calls=3, total_tokens=6000,
    )
    t_agent = make_trace(metadata={"chatMode": "agent"}, summary=dict(common))
    t_norm = make_trace(metadata={"chatMode": "normal"}, summary=dict(common))
    _, dims_a = compute_reward_signals(t_agent)
    _, dims_n = compute_reward_signals(t_norm)
    a = {d.name: d.value for d in dims_a}
    n = {d.name: d.value for d in dims_n}
    assert a["tool_call_reliability"] == -0.5
    assert n["tool_call_reliability"] == -1.0
    # tool count 10: agent <= 15 (good) but > excellent 8 -> 0.3; normal: 10 <= fair(10) but > good(6) -> -0.3
    assert a["tool_call_efficiency"] == 0.3
    assert n["tool_call_efficiency"] == -0.3
    # llm calls 3: agent thr 3 -> 1.0; normal thr 1 -> 1-0.8=0.2
    assert a["response_efficiency"] == pytest.approx(1.0)
    assert n["response_efficiency"] == pytest.approx(0.19999999999999996)
    # tokens 6000: agent > excellent 5000 -> 0.5; normal > good 5000 -> 0.0
    assert a["token_efficiency"] == 0.5
    assert n["token_efficiency"] == 0.0


def test_response_efficiency_floor():
    t = make_trace(summary=dict(total_llm_calls=20))
    _, dims = compute_reward_signals(t)
    assert {d.name: d.value for d in dims}["response_efficiency"] == -1.0


def test_conversation_efficiency_tiers():
    for turns, expected in [(2, 1.0), (3, 0.3), (5, -0.3), (7, -0.8)]:
        t = make_trace(summary=dict())
        add_msgs(t, turns, turns)
        _, dims = compute_reward_signals(t)
        assert {d.name: d.value for d in dims}["conversation_efficiency"] == expected, turns


def test_duration_efficiency_tiers():
    for avg_ms, expected in [(500, 1.0), (2000, 0.5), (5000, 0.0), (20000, -0.5)]:
        t = make_trace(summary=dict(total_tool_calls=1, tool_calls_succeeded=1,
                                    total_tool_duration_ms=float(avg_ms)))
        _, dims = compute_reward_signals(t)
        assert {d.name: d.value for d in dims}["tool_duration_efficiency"] == expected, avg_ms


def test_reward_via_collector_end_to_end():
    c = TraceCollector()
    tid = c.start_trace("thread-1", {"chatMode": "normal"})
    c.record_user_message("thread-1", 0, "fix the bug")
    c.record_llm_call("thread-1", 1, input_tokens=1000, output_tokens=500, duration=800)
    c.record_tool_call("thread-1", 1, tool_name="read_file", tool_success=True, duration=120)
    c.record_assistant_message("thread-1", 1, "done")
    c.end_trace(tid)
    c.record_user_feedback("thread-1", 1, "good")
    trace = c.get_all_traces()[0]
    assert trace.summary.final_reward is not None
    assert trace.summary.user_feedback == "good"
    by_name = {d.name: d.value for d in trace.summary.reward_dimensions}
    assert by_name["user_feedback"] == 1.0
    assert by_name["tool_success_rate"] == 1.0
    stats = c.get_stats()
    assert stats["totalTraces"] == 1
    assert stats["goodFeedbacks"] == 1
    assert stats["toolSuccessRate"] == 1.0
