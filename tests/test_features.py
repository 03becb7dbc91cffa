"""Feature services: settings, autocomplete/FIM, edit prediction, snapshots, SCM."""

import subprocess

from senweaver_amd.apo.optimizer import StubBackend
from senweaver_amd.features import (
    AutocompleteService,
    EditPredictionService,
    FileSnapshotService,
    SCMService,
    SettingsService,
    build_fim_prompt,
    find_diffs,
)
from senweaver_amd.storage import MemoryStorage


def test_settings_roundtrip():
    st = MemoryStorage()
    s = SettingsService(st)
    s.set_chat_mode("designer")
    s.set_model_selection("Chat", "local", "llama-3-70b")
    s.set_auto_approve("edits", True)
    s2 = SettingsService(st)
    assert s2.global_settings.chat_mode == "designer"
    assert s2.model_selection_of_feature["Chat"].model_name == "llama-3-70b"
    assert s2.global_settings.auto_approve == {"edits": True}


def test_fim_prompt_and_cache():
    backend = StubBackend()
    ac = AutocompleteService(backend)
    out1 = ac.complete("def add(a, b):\n    return ", "\n\nprint(add(1,2))")
    out2 = ac.complete("def add(a, b):\n    return ", "\n\nprint(add(1,2))")
    assert out1 == out2
    assert len(backend.generate_calls) == 1  # second hit came from the LRU
    assert "<|fim_prefix|>" in backend.generate_calls[0]
    assert "<|fim_middle|>" in backend.generate_calls[0]
    p = build_fim_prompt("x" * 5000, "y" * 5000)
    assert len(p) < 3000  # context caps applied


def test_edit_prediction_cache():
    backend = StubBackend()
    ep = EditPredictionService(backend)
    ep.record_edit("a.py", "x = 1", "x = 2")
    p1 = ep.predict_next_edit("a.py", "y = 1\n")
    p2 = ep.predict_next_edit("a.py", "y = 1\n")
    assert p1 == p2 and len(backend.generate_calls) == 1


def test_snapshots_and_restore(tmp_path):
    f = tmp_path / "code.py"
    f.write_text("version = 1\n")
    svc = FileSnapshotService(str(tmp_path))
    snap = svc.ensure_before_state("code.py", turn_id="t1")
    assert snap is not None
    assert svc.ensure_before_state("code.py", turn_id="t1") is None  # once per turn
    f.write_text("version = 2\nextra = True\n")
    hunks = svc.diff_against_current(snap.id)
    assert hunks and hunks[0].orig_lines == ["version = 1"]
    assert svc.restore(snap.id)
    assert f.read_text() == "version = 1\n"


def test_find_diffs():
    h = find_diffs("a\nb\nc", "a\nB\nc\nd")
    assert any(x.orig_lines == ["b"] and x.new_lines == ["B"] for x in h)


def test_scm_commit_message(tmp_path):
    subprocess.run(["git", "init", "-q", str(tmp_path)], check=True)
    (tmp_path / "f.txt").write_text("hello\n")
    subprocess.run(["git", "-C", str(tmp_path), "add", "."], check=True)
    svc = SCMService(StubBackend(), str(tmp_path))
    diff = svc.collect_diff(staged=True)
    assert "hello" in diff
    msg = svc.generate_commit_message()
    assert msg  # produced by the backend


# ---------------- AIRegexService ----------------

class _ScriptedBackend:
    def __init__(self, reply):
        self.reply = reply

    def generate(self, prompt, max_new_tokens=64):
        if isinstance(self.reply, Exception):
            raise self.reply
        return self.reply


def test_airegex_valid_generation():
    from senweaver_amd.features.airegex import AIRegexService
    svc = AIRegexService(_ScriptedBackend(r"\d{3}-\d{4}"))
    r = svc.generate("a phone number like 555-1234")
    assert r.pattern == r"\d{3}-\d{4}" and not r.is_fallback


def test_airegex_slash_delimited_and_fenced():
    from senweaver_amd.features.airegex import AIRegexService
    svc = AIRegexService(_ScriptedBackend("```\n/foo+bar/gi\n```"))
    r = svc.generate("foo then bar")
    assert r.pattern == "foo+bar" and not r.is_fallback


def test_airegex_invalid_falls_back_literal():
    from senweaver_amd.features.airegex import AIRegexService
    import re
    svc = AIRegexService(_ScriptedBackend("([unclosed"))
    r = svc.generate("a (weird) thing")
    assert r.is_fallback and re.search(r.pattern, "a (weird) thing")


def test_airegex_offline_fallback():
    from senweaver_amd.features.airegex import AIRegexService
    svc = AIRegexService(None)
    r = svc.generate("c++ code")
    assert r.is_fallback
    import re
    assert re.search(r.pattern, "some c++ code here")


# ---------------- MetricsPoller ----------------

def test_metrics_poller_heartbeat():
    from senweaver_amd.utils.observability import MetricsPoller, MetricsService
    ms = MetricsService()
    ms.capture("some_event", {"a": 1})
    p = MetricsPoller(ms, interval_s=9999, extra_props=lambda: {"gpu": 1})
    p.poll_once()
    p.poll_once()
    beats = [e for e in ms.events if e["event"] == "metrics_poll_heartbeat"]
    assert len(beats) == 2
    assert beats[0]["poll"] == 0 and beats[1]["poll"] == 1
    assert beats[0]["gpu"] == 1 and beats[0]["totalEvents"] >= 1


def test_trace_range_records():
    from senweaver_amd.utils.observability import PerformanceMonitor, trace_range
    mon = PerformanceMonitor(enabled=True)
    with trace_range("messageTrimming", mon):
        pass
    assert "messageTrimming" in mon.summary()


def test_model_capabilities_registry():
    """modelCapabilities.ts rebuild: record lookup, substring fallback, and
    the reserved-output-space default semantics."""
    from senweaver_amd.models.capabilities import (
        DEFAULT_RESERVED_OUTPUT_TOKEN_SPACE, ModelCapabilities,
        get_context_window, get_model_capabilities,
        get_reserved_output_token_space,
    )

    caps = get_model_capabilities("llama-3-8b")
    assert caps.contextWindow == 8192 and caps.supportsFIM
    assert caps.specialToolFormat is None  # XML tool calls in agent mode
    assert caps.supportsSystemMessage == "system-role"
    # substring family fallback, like the reference's recognizable-model map
    assert get_model_capabilities("my-llama-finetune").contextWindow == 8192
    assert get_model_capabilities("mixtral-custom").contextWindow == 32768
    # unknown -> default record; None reserved space -> 4096 default
    unk = get_model_capabilities("who-knows")
    assert unk.contextWindow == 4096 and unk.reservedOutputTokenSpace is None
    assert (get_reserved_output_token_space("who-knows")
            == DEFAULT_RESERVED_OUTPUT_TOKEN_SPACE)
    assert get_reserved_output_token_space("llama-3-8b") == 2048
    assert get_context_window("tiny-debug") == 2048
    assert isinstance(caps, ModelCapabilities)
