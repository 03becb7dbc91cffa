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
