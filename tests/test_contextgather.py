"""ContextGatheringService rebuild (VERDICT r01 missing #5).

Reference: browser/contextGatheringService.ts — NUM_LINES=3 padding,
MAX_SNIPPET_LINES=7 cap, comment/blank cleaning, visited-interval dedup,
nearby + container + symbol-definition gathering.
"""

import textwrap

import pytest

from senweaver_amd.features.contextgather import (
    ContextGatheringService, MAX_SNIPPET_LINES,
)


@pytest.fixture()
def ws(tmp_path):
    (tmp_path / "util.py").write_text(textwrap.dedent("""\
        def helper_fn(x):
            return x * 2

        THRESHOLD_VALUE = 42
        """))
    (tmp_path / "main.py").write_text(textwrap.dedent("""\
        import os

        def outer_container():
            a = 1
            b = helper_fn(a)
            c = b + THRESHOLD_VALUE
            return c

        def unrelated():
            pass
        """))
    return tmp_path


def test_nearby_and_definition_snippets(ws):
    svc = ContextGatheringService(str(ws))
    snips = svc.update_cache("main.py", 5)  # on the helper_fn call
    joined = "\n---\n".join(snips)
    assert "helper_fn(a)" in joined              # nearby snippet
    assert "def helper_fn(x):" in joined         # cross-file definition
    assert "THRESHOLD_VALUE = 42" in joined      # assignment definition
    assert svc.get_cached_snippets() == snips


def test_container_gathered(ws):
    svc = ContextGatheringService(str(ws))
    snips = svc.update_cache("main.py", 6)
    assert any("def outer_container" in s for s in snips)


def test_snippet_cap_and_cleaning(tmp_path):
    body = "\n".join([f"line{i} = {i}" for i in range(40)])
    (tmp_path / "big.py").write_text("def f():\n" + body + "\n\n\n//\n")
    svc = ContextGatheringService(str(tmp_path))
    snips = svc.update_cache("big.py", 20)
    for s in snips:
        assert len(s.split("\n")) <= MAX_SNIPPET_LINES
        assert "" not in s.split("\n")          # blank lines cleaned
        assert "//" not in s                     # comment-only line cleaned


def test_overlap_dedup(tmp_path):
    # two symbols defined on adjacent lines: their ranges overlap after the
    # +-3 padding -> only one snippet is gathered for that interval
    (tmp_path / "m.py").write_text(
        "first_sym = 1\nsecond_sym = 2\n\n\n\n\n\n\n\n"
        "x = first_sym + second_sym\n")
    svc = ContextGatheringService(str(tmp_path))
    snips = svc.update_cache("m.py", 10)
    defs = [s for s in snips if "first_sym = 1" in s]
    assert len(defs) == 1


def test_open_file_overrides_disk(tmp_path):
    (tmp_path / "f.py").write_text("on_disk = 1\n")
    svc = ContextGatheringService(str(tmp_path))
    svc.open_file("f.py", "in_memory_version = 99\n")
    snips = svc.update_cache("f.py", 1)
    assert any("in_memory_version" in s for s in snips)
