"""File snapshot service — before-edit state capture and rollback.

Capability-compatible with the reference's FileSnapshotService family
(browser/fileSnapshotService.ts + fileSnapshotTypes.ts): snapshots are taken
once per file per agent turn before the first edit
(chatThreadService.ts:1062-1069 _ensureFileBeforeStateIsSaved), kept as a
per-file history stack, restorable, and diffable (the findDiffs helper).
"""

from __future__ import annotations

import difflib
import os
import time
import uuid
from dataclasses import dataclass
from typing import Dict, List, Optional


@dataclass
class FileSnapshot:
    id: str
    uri: str
    content: str
    taken_at: float
    turn_id: Optional[str] = None


@dataclass
class DiffHunk:
    """A contiguous change block (browser/helpers/findDiffs.ts analog)."""
    orig_start: int   # 1-based line numbers
    orig_lines: List[str]
    new_start: int
    new_lines: List[str]


def find_diffs(original: str, modified: str) -> List[DiffHunk]:
    sm = difflib.SequenceMatcher(a=original.split("\n"), b=modified.split("\n"),
                                 autojunk=False)
    hunks: List[DiffHunk] = []
    for tag, i1, i2, j1, j2 in sm.get_opcodes():
        if tag == "equal":
            continue
        hunks.append(DiffHunk(
            orig_start=i1 + 1, orig_lines=original.split("\n")[i1:i2],
            new_start=j1 + 1, new_lines=modified.split("\n")[j1:j2]))
    return hunks


class FileSnapshotService:
    def __init__(self, workspace_root: str, max_snapshots_per_file: int = 20) -> None:
        self.root = os.path.abspath(workspace_root)
        self._stacks: Dict[str, List[FileSnapshot]] = {}
        self._turn_snapshotted: Dict[str, set] = {}
        self._max = max_snapshots_per_file

    def _read(self, uri: str) -> str:
        path = os.path.join(self.root, uri) if not os.path.isabs(uri) else uri
        try:
            with open(path, "r", encoding="utf-8", errors="replace") as f:
                return f.read()
        except OSError:
            return ""

    def ensure_before_state(self, uri: str, turn_id: str) -> Optional[FileSnapshot]:
        """Snapshot once per file per turn (the agent-loop contract)."""
        taken = self._turn_snapshotted.setdefault(turn_id, set())
        if uri in taken:
            return None
        taken.add(uri)
        return self.take_snapshot(uri, turn_id)

    def take_snapshot(self, uri: str, turn_id: Optional[str] = None) -> FileSnapshot:
        snap = FileSnapshot(id=str(uuid.uuid4()), uri=uri, content=self._read(uri),
                            taken_at=time.time(), turn_id=turn_id)
        stack = self._stacks.setdefault(uri, [])
        stack.append(snap)
        if len(stack) > self._max:
            del stack[0]
        return snap

    def history(self, uri: str) -> List[FileSnapshot]:
        return list(self._stacks.get(uri, []))

    def restore(self, snapshot_id: str) -> bool:
        for uri, stack in self._stacks.items():
            for snap in stack:
                if snap.id == snapshot_id:
                    path = os.path.join(self.root, uri) if not os.path.isabs(uri) else uri
                    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
                    with open(path, "w", encoding="utf-8") as f:
                        f.write(snap.content)
                    return True
        return False

    def diff_against_current(self, snapshot_id: str) -> Optional[List[DiffHunk]]:
        for uri, stack in self._stacks.items():
            for snap in stack:
                if snap.id == snapshot_id:
                    return find_diffs(snap.content, self._read(uri))
        return None
