"""Golden-format regression tests: the serialized JSON must stay stable.

These lock the persistence/export shapes byte-for-byte (deterministic clock
and uuid), so any accidental format drift from the reference's storage
contract fails loudly.
"""

import json

from senweaver_amd.apo import APOService, VersionedPromptTemplate
from senweaver_amd.storage import (
    APO_BEAM_STATE_KEY,
    APO_SEGMENTS_KEY,
    MemoryStorage,
    TRACE_STORAGE_KEY,
)
from senweaver_amd.trace import TraceCollector
from senweaver_amd.utils.jsonutil import js_stringify


def deterministic_services():
    t = {"now": 1700000000000}

    def clock():
        t["now"] += 10
        return t["now"]

    n = {"i": 0}

    def uuid_fn():
        n["i"] += 1
        return f"00000000-0000-4000-8000-{n['i']:012d}"

    storage = MemoryStorage()
    tc = TraceCollector(storage=storage, clock=clock, uuid_fn=uuid_fn)
    apo = APOService(tc, storage=storage, clock=clock, uuid_fn=uuid_fn)
    return storage, tc, apo


GOLDEN_TRACE = (
    "[{\"id\":\"00000000-0000-4000-8000-000000000001\",\"threadId\":\"th\",\"startTime\":1700000000020,\"endTime\":17"
    "00000000060,\"spans\":[{\"id\":\"00000000-0000-4000-8000-000000000002\",\"traceId\":\"00000000-0000-4000-8000"
    "-000000000001\",\"threadId\":\"th\",\"messageIdx\":0,\"type\":\"user_message\",\"timestamp\":1700000000030,\"data\""
    ":{\"contentPreview\":\"hello\",\"contentLength\":5}},{\"id\":\"00000000-0000-4000-8000-000000000003\",\"traceId"
    "\":\"00000000-0000-4000-8000-000000000001\",\"threadId\":\"th\",\"messageIdx\":1,\"type\":\"tool_call\",\"timestam"
    "p\":1700000000040,\"duration\":25,\"data\":{\"toolName\":\"read_file\",\"toolParams\":\"\",\"toolResult\":\"ok\",\"too"
    "lSuccess\":true}},{\"id\":\"00000000-0000-4000-8000-000000000004\",\"traceId\":\"00000000-0000-4000-8000-000"
    "000000001\",\"threadId\":\"th\",\"messageIdx\":2,\"type\":\"assistant_message\",\"timestamp\":1700000000050,\"data"
    "\":{\"contentPreview\":\"done\",\"contentLength\":4}},{\"id\":\"00000000-0000-4000-8000-000000000005\",\"traceId"
    "\":\"00000000-0000-4000-8000-000000000001\",\"threadId\":\"th\",\"messageIdx\":2,\"type\":\"user_feedback\",\"time"
    "stamp\":1700000000080,\"data\":{\"feedback\":\"good\"}}],\"metadata\":{\"chatMode\":\"normal\"},\"summary\":{\"total"
    "LLMCalls\":0,\"totalToolCalls\":1,\"totalTokens\":0,\"userFeedback\":\"good\",\"hasErrors\":false,\"toolCallsSuc"
    "ceeded\":1,\"toolCallsFailed\":0,\"toolCallsByName\":{\"read_file\":{\"total\":1,\"succeeded\":1,\"failed\":0}},\""
    "totalToolDurationMs\":25,\"finalReward\":1,\"rewardDimensions\":[{\"name\":\"user_feedback\",\"value\":1},{\"nam"
    "e\":\"task_completion\",\"value\":1},{\"name\":\"tool_success_rate\",\"value\":1},{\"name\":\"tool_call_reliabilit"
    "y\",\"value\":1},{\"name\":\"tool_call_efficiency\",\"value\":1},{\"name\":\"tool_duration_efficiency\",\"value\":1"
    "},{\"name\":\"conversation_efficiency\",\"value\":1}]}}]"
)


def test_trace_storage_golden_bytes():
    storage, tc, _ = deterministic_services()
    tid = tc.start_trace("th", {"chatMode": "normal"})
    tc.record_user_message("th", 0, "hello")
    tc.record_tool_call("th", 1, tool_name="read_file", tool_success=True,
                        tool_result="ok", duration=25)
    tc.record_assistant_message("th", 2, "done")
    tc.end_trace(tid)
    tc.record_user_feedback("th", 2, "good")
    tc.flush()
    stored = storage.get(TRACE_STORAGE_KEY)
    assert stored == GOLDEN_TRACE
    data = json.loads(stored)
    assert abs(data[0]["summary"]["finalReward"] - 1) < 1e-12


def test_beam_state_storage_golden():
    storage, tc, apo = deterministic_services()
    state = apo.ensure_beam_state()
    apo.apply_beam_update({
        "beam": [VersionedPromptTemplate("v0", "- rule", 0.5, 1700000000000)],
        "bestPrompt": VersionedPromptTemplate("v0", "- rule", 0.5, 1700000000000),
        "bestScore": 0.5,
        "round": 1,
    })
    apo.flush()
    raw = json.loads(storage.get(APO_BEAM_STATE_KEY))
    assert set(raw.keys()) == {"currentRound", "totalRounds", "beam", "historyBestPrompt",
                               "historyBestScore", "versionCounter", "startedAt",
                               "lastUpdatedAt"}
    assert raw["beam"][0] == {"version": "v0", "content": "- rule", "score": 0.5,
                              "createdAt": 1700000000000}
    segs = json.loads(storage.get(APO_SEGMENTS_KEY))
    assert segs[0]["category"] == "core_behavior"
    assert segs[0]["isOptimized"] is True


def test_js_stringify_quirks():
    # integral floats print as ints; None -> null; unicode unescaped
    assert js_stringify({"a": 1.0, "b": None, "c": "好"}) == '{"a":1,"b":null,"c":"好"}'
    assert js_stringify(float("nan")) == "null"
    assert js_stringify([0.5, 2.0]) == "[0.5,2]"


def test_file_storage_restart_resume(tmp_path):
    """The reference's durability contract (README:143 'restart loses
    nothing'): beam state + segments + traces written through FileStorage
    reload in fresh service instances from the same path."""
    from senweaver_amd.storage import FileStorage

    path = str(tmp_path / "state.json")

    def services(storage):
        t = {"now": 1700000000000}

        def clock():
            t["now"] += 10
            return t["now"]
        n = {"i": 0}

        def uuid_fn():
            n["i"] += 1
            return f"00000000-0000-4000-8000-{n['i']:012d}"
        tc = TraceCollector(storage=storage, clock=clock, uuid_fn=uuid_fn)
        return tc, APOService(tc, storage=storage, clock=clock, uuid_fn=uuid_fn)

    s1 = FileStorage(path)
    tc1, apo1 = services(s1)
    tid = tc1.start_trace("th", {"chatMode": "agent"})
    tc1.record_user_message("th", 0, "persist me")
    tc1.end_trace(tid)
    tc1.flush()
    apo1.ensure_beam_state()
    apo1.apply_beam_update({
        "beam": [VersionedPromptTemplate("v7", "- durable rule", 0.9, 1700000000000)],
        "bestPrompt": VersionedPromptTemplate("v7", "- durable rule", 0.9, 1700000000000),
        "bestScore": 0.9,
        "round": 2,
    })
    apo1.flush()
    s1.flush()

    # fresh process: everything reloads
    s2 = FileStorage(path)
    tc2, apo2 = services(s2)
    assert len(tc2.get_all_traces()) == 1
    state = apo2.get_beam_state()
    assert state is not None and state.current_round == 2
    assert state.history_best_prompt.content == "- durable rule"
    assert apo2.get_optimized_rules()  # optimized segment survived

    # corrupted file: silent tolerant load (reference storage-load path)
    with open(path, "w") as f:
        f.write("{not json")
    s3 = FileStorage(path)
    assert s3.get("senweaver.apo.beamState") is None
