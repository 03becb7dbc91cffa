"""StreamingEditSession / EditCodeService (VERDICT r01 partial #13).

Reference: browser/editCodeService.ts — streamed search/replace application
with per-zone Accept/Reject and whole-session revert.
"""

import pytest

from senweaver_amd.features.editcode import EditCodeService, StreamingEditSession
from senweaver_amd.utils.codeextract import DIVIDER, FINAL, ORIGINAL

DOC = "alpha\nbeta\ngamma\ndelta\n"


def _block(orig, final):
    return f"{ORIGINAL}\n{orig}\n{DIVIDER}\n{final}\n{FINAL}\n"


def test_blocks_apply_as_stream_completes():
    sess = StreamingEditSession(DOC)
    full = _block("beta", "BETA") + _block("delta", "DELTA")
    events = []
    sess._on_zone_change = events.append
    # feed the stream in growing prefixes (cumulative contract)
    for cut in range(0, len(full) + 1, 7):
        sess.on_stream_text(full[:cut])
    sess.on_stream_final(full)
    assert sess.current_text == "alpha\nBETA\ngamma\nDELTA\n"
    assert [z.state for z in sess.zones] == ["applied", "applied"]
    assert len(events) == 2  # one event per completed zone, no re-fires


def test_partial_block_not_applied_early():
    sess = StreamingEditSession(DOC)
    sess.on_stream_text(f"{ORIGINAL}\nbeta\n{DIVIDER}\nBET")
    assert sess.current_text == DOC
    assert sess.zones == []


def test_reject_restores_zone_text():
    sess = StreamingEditSession(DOC)
    sess.on_stream_final(_block("beta", "BETA") + _block("gamma", "GAMMA"))
    sess.reject(0)
    assert sess.current_text == "alpha\nbeta\nGAMMA\ndelta\n"
    assert sess.zones[0].state == "rejected"
    sess.accept(1)
    assert sess.zones[1].state == "accepted"
    with pytest.raises(ValueError):
        sess.reject(1)


def test_reject_all_reverts_everything():
    sess = StreamingEditSession(DOC)
    sess.on_stream_final(_block("alpha", "A") + _block("delta", "D"))
    sess.accept_all()
    sess.reject_all()
    assert sess.current_text == DOC
    assert all(z.state == "rejected" for z in sess.zones)


def test_missing_original_marks_failed_zone():
    sess = StreamingEditSession(DOC)
    sess.on_stream_final(_block("not-there", "x") + _block("beta", "B"))
    assert sess.zones[0].state == "failed"
    assert sess.zones[1].state == "applied"  # later blocks still apply
    assert "B" in sess.current_text


def test_service_writes_back_on_close(tmp_path):
    f = tmp_path / "doc.txt"
    f.write_text(DOC)
    svc = EditCodeService(
        read_file=lambda uri: (tmp_path / uri).read_text(),
        write_file=lambda uri, text: (tmp_path / uri).write_text(text))
    sess = svc.start_session("doc.txt")
    sess.on_stream_final(_block("gamma", "G"))
    assert svc.get_session("doc.txt") is sess
    out = svc.close_session("doc.txt")
    assert "G" in out and f.read_text() == out
    assert svc.get_session("doc.txt") is None
