"""SCM service — AI git commit-message generation from diffs.

Capability-compatible with the reference's SenweaverSCMService
(browser/senweaverSCMService.ts + electron-main/senweaverSCMMainService.ts):
collects the staged/working diff and asks the model for a conventional
commit message.  Runs `git` locally instead of an IPC hop.
"""

from __future__ import annotations

import subprocess
from typing import Optional

MAX_DIFF_CHARS = 12_000

COMMIT_PROMPT = """Write a concise git commit message for the following diff.
Rules: one summary line under 72 characters (imperative mood), then an
optional short body. Return only the commit message.

## Diff
{diff}
"""


class SCMService:
    def __init__(self, backend, repo_root: str) -> None:
        self._backend = backend
        self.root = repo_root

    def collect_diff(self, staged: bool = True) -> str:
        args = ["git", "-C", self.root, "diff"]
        if staged:
            args.append("--cached")
        try:
            out = subprocess.run(args, capture_output=True, text=True, timeout=30).stdout
        except (OSError, subprocess.TimeoutExpired):
            return ""
        if len(out) > MAX_DIFF_CHARS:
            out = out[:MAX_DIFF_CHARS] + "\n...[diff truncated]"
        return out

    def generate_commit_message(self, diff: Optional[str] = None,
                                max_new_tokens: int = 64) -> str:
        if diff is None:
            diff = self.collect_diff(staged=True) or self.collect_diff(staged=False)
        if not diff:
            return "chore: no changes"
        return self._backend.generate(COMMIT_PROMPT.format(diff=diff),
                                      max_new_tokens=max_new_tokens).strip()
