"""EditCodeService — streaming search/replace application with per-block
accept/reject/revert.

Rebuild of browser/editCodeService.ts (2636 LoC): the reference streams an
LLM's search/replace blocks into an editor, materializing each diff zone as
it completes and offering Accept/Reject widgets per zone plus accept-all /
reject-all.  The engine analog keeps the same lifecycle without the Monaco
widgets: a StreamingEditSession consumes the token stream incrementally
(utils/codeextract.extract_search_replace_blocks — the parser whose state
machine guarantees monotone progress), tracks one DiffZone per block
(pending -> applied; accepted/rejected), applies completed blocks to a
working copy as they close, and supports per-zone reject (restores that
zone's original text) and whole-session revert.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, List, Optional

from ..utils.codeextract import extract_search_replace_blocks


@dataclass
class DiffZone:
    index: int
    orig: str
    final: str
    state: str = "pending"     # pending | applied | accepted | rejected | failed
    error: Optional[str] = None


class StreamingEditSession:
    """One streamed edit over one document.

    Feed cumulative LLM text with ``on_stream_text`` (the transport's
    cumulative onText contract); completed blocks apply immediately to the
    working text.  Then accept/reject zones and read ``current_text``.
    """

    def __init__(self, original_text: str,
                 on_zone_change: Optional[Callable[[DiffZone], None]] = None
                 ) -> None:
        self.original_text = original_text
        self.current_text = original_text
        self.zones: List[DiffZone] = []
        self._applied_count = 0
        self._on_zone_change = on_zone_change or (lambda z: None)
        self._stream_done = False

    # ---- streaming input ----
    def on_stream_text(self, full_text: str) -> None:
        """Cumulative stream callback: applies newly COMPLETED blocks."""
        blocks = extract_search_replace_blocks(full_text)
        for i in range(self._applied_count, len(blocks)):
            b = blocks[i]
            if b.state != "done":
                break  # still streaming this block
            zone = DiffZone(index=i, orig=b.orig, final=b.final)
            if b.orig and b.orig in self.current_text:
                self.current_text = self.current_text.replace(b.orig, b.final, 1)
                zone.state = "applied"
            elif not b.orig:
                zone.state = "failed"
                zone.error = "empty ORIGINAL block"
            else:
                zone.state = "failed"
                zone.error = f"ORIGINAL not found: {b.orig[:60]!r}"
            self.zones.append(zone)
            self._applied_count += 1
            self._on_zone_change(zone)

    def on_stream_final(self, full_text: str) -> None:
        self.on_stream_text(full_text)
        self._stream_done = True

    # ---- accept / reject (the per-zone widgets) ----
    def accept(self, index: int) -> None:
        z = self._zone(index)
        if z.state == "applied":
            z.state = "accepted"
            self._on_zone_change(z)

    def reject(self, index: int) -> None:
        """Restore this zone's original text in the working copy."""
        z = self._zone(index)
        if z.state in ("applied",):
            if z.final and z.final in self.current_text:
                self.current_text = self.current_text.replace(z.final, z.orig, 1)
                z.state = "rejected"
            elif not z.final:
                # pure deletion was applied; re-insert is ambiguous without
                # position info — treat as failed reject
                z.error = "cannot locate applied edit to revert"
            else:
                z.error = "cannot locate applied edit to revert"
            self._on_zone_change(z)
        elif z.state == "accepted":
            raise ValueError("zone already accepted")

    def accept_all(self) -> None:
        for z in self.zones:
            if z.state == "applied":
                z.state = "accepted"
                self._on_zone_change(z)

    def reject_all(self) -> None:
        """Whole-session revert to the original document."""
        self.current_text = self.original_text
        for z in self.zones:
            if z.state in ("applied", "accepted"):
                z.state = "rejected"
                self._on_zone_change(z)

    # ---- status ----
    def pending_zones(self) -> List[DiffZone]:
        return [z for z in self.zones if z.state == "applied"]

    def is_done(self) -> bool:
        return self._stream_done

    def _zone(self, index: int) -> DiffZone:
        for z in self.zones:
            if z.index == index:
                return z
        raise KeyError(f"no diff zone {index}")


class EditCodeService:
    """Session registry: one streaming session per (uri); writes back on
    close (the editCodeService 'apply to model' step)."""

    def __init__(self, read_file: Callable[[str], str],
                 write_file: Callable[[str, str], None]) -> None:
        self._read = read_file
        self._write = write_file
        self._sessions: dict = {}

    def start_session(self, uri: str) -> StreamingEditSession:
        sess = StreamingEditSession(self._read(uri))
        self._sessions[uri] = sess
        return sess

    def get_session(self, uri: str) -> Optional[StreamingEditSession]:
        return self._sessions.get(uri)

    def close_session(self, uri: str, write: bool = True) -> str:
        sess = self._sessions.pop(uri)
        if write:
            self._write(uri, sess.current_text)
        return sess.current_text
