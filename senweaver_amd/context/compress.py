"""History compression + smart-context management.

Rebuilds the reference's long-context cascade (SURVEY.md §5.7 — the
reference scales sequence length *down*, not out):

- per-message compression of old history with per-tool compaction
  strategies (browser/convertToLLMMessageService.ts:960-1100:
  KEEP_RECENT_COUNT=10; read_file -> identifier summary; run_command ->
  head+tail),
- tool-output pruning at 55% occupancy with a 20k-token protection window
  (common/smartContextManager.ts:19-101, EnhancedContextManager :684),
- the token estimator (3.5 chars/token, cached).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

CHARS_PER_TOKEN = 3.5
KEEP_RECENT_COUNT = 10

# SMART_CONTEXT_CONFIG (smartContextManager.ts:19-101)
SMART_CONTEXT_CONFIG = {
    "OVERFLOW_THRESHOLD": 0.55,          # prune when occupancy exceeds 55%
    "PRUNE_PROTECT_TOKENS": 20_000,      # most recent tokens protected
    "LARGE_OUTPUT_THRESHOLD": 50_000,    # chars: outputs above this prune first
    "COMPRESSION_MIN_SAVING": 500,       # don't bother below this
    "DEFAULT_CONTEXT_LIMIT": 128_000,
}

MODEL_CONTEXT_LIMITS = {
    # capability table subset (modelCapabilities.ts / smartContextManager.ts:75-100)
    "llama-3-8b": 8192,
    "llama-3-70b": 8192,
    "mixtral-8x7b": 32768,
    "default": 128_000,
}


class TokenEstimator:
    """3.5 chars/token with memoization (smartContextManager.ts:137)."""

    def __init__(self) -> None:
        self._cache: Dict[int, int] = {}

    def estimate(self, text: str) -> int:
        key = id(text) if len(text) > 4096 else hash(text)
        got = self._cache.get(key)
        if got is not None:
            return got
        n = int(len(text) / CHARS_PER_TOKEN) + 1
        if len(self._cache) > 4096:
            self._cache.clear()
        self._cache[key] = n
        return n


_IDENT_RE = re.compile(r"^\s*(?:def|class|function|const|let|var|export|public|private|interface|struct|fn)\b.*", re.M)


def compact_tool_output(tool_name: Optional[str], content: str) -> str:
    """Per-tool-name compaction strategies (convertToLLMMessageService.ts:960-1030)."""
    if len(content) <= 1000:
        return content
    if tool_name == "read_file":
        # identifier summary: keep declaration lines
        idents = _IDENT_RE.findall(content)[:40]
        header = content[:300]
        return (f"{header}\n...[file compacted: {len(content)} chars; declarations:]\n"
                + "\n".join(idents))
    if tool_name in ("run_command", "run_persistent_command"):
        return content[:400] + f"\n...[{len(content) - 800} chars omitted]...\n" + content[-400:]
    if tool_name in ("search_for_files", "search_pathnames_only", "ls_dir"):
        lines = content.split("\n")
        if len(lines) > 30:
            return "\n".join(lines[:30]) + f"\n...[{len(lines) - 30} more results omitted]"
        return content
    return content[:800] + "...[compacted]"


@dataclass
class CompressibleMessage:
    role: str
    content: str
    tool_name: Optional[str] = None


def compress_old_messages(messages: List[CompressibleMessage],
                          keep_recent: int = KEEP_RECENT_COUNT) -> List[CompressibleMessage]:
    """Compact everything except the most recent ``keep_recent`` messages."""
    cutoff = max(0, len(messages) - keep_recent)
    out: List[CompressibleMessage] = []
    for i, m in enumerate(messages):
        if i >= cutoff or m.role == "user":
            out.append(m)
        elif m.role == "tool":
            out.append(CompressibleMessage(m.role, compact_tool_output(m.tool_name, m.content), m.tool_name))
        else:
            c = m.content if len(m.content) <= 1000 else m.content[:1000] + "...[compressed]"
            out.append(CompressibleMessage(m.role, c, m.tool_name))
    return out


def needs_compression(messages: List[CompressibleMessage], context_limit: int) -> bool:
    """messageCompressor.ts:284 — compress when estimated tokens exceed half."""
    est = sum(int(len(m.content) / CHARS_PER_TOKEN) for m in messages)
    return est > context_limit * 0.5


@dataclass
class CompactionState:
    pruned_tool_ids: Set[int] = field(default_factory=set)


class EnhancedContextManager:
    """OpenCode-style tool-output pruning (smartContextManager.ts:684+):
    when occupancy passes OVERFLOW_THRESHOLD, prune the oldest/largest tool
    outputs outside the 20k-token protection window."""

    def __init__(self, context_limit: int = SMART_CONTEXT_CONFIG["DEFAULT_CONTEXT_LIMIT"]):
        self.context_limit = context_limit
        self.compaction_state = CompactionState()
        self.estimator = TokenEstimator()

    def is_tool_pruned(self, msg_id: int) -> bool:
        return msg_id in self.compaction_state.pruned_tool_ids

    def maybe_prune(self, messages: List[CompressibleMessage]) -> List[CompressibleMessage]:
        total = sum(self.estimator.estimate(m.content) for m in messages)
        if total <= self.context_limit * SMART_CONTEXT_CONFIG["OVERFLOW_THRESHOLD"]:
            return messages
        # walk from the end, protecting the newest 20k tokens
        protected = 0
        protect_until = len(messages)
        for i in range(len(messages) - 1, -1, -1):
            protected += self.estimator.estimate(messages[i].content)
            if protected > SMART_CONTEXT_CONFIG["PRUNE_PROTECT_TOKENS"]:
                protect_until = i + 1
                break
        else:
            protect_until = 0
        out = list(messages)
        # prune large tool outputs first, then any tool output, oldest first
        candidates = [i for i in range(protect_until) if out[i].role == "tool"
                      and i not in self.compaction_state.pruned_tool_ids]
        candidates.sort(key=lambda i: (-(len(out[i].content) >
                                         SMART_CONTEXT_CONFIG["LARGE_OUTPUT_THRESHOLD"]), i))
        for i in candidates:
            if total <= self.context_limit * SMART_CONTEXT_CONFIG["OVERFLOW_THRESHOLD"]:
                break
            saved = self.estimator.estimate(out[i].content)
            out[i] = CompressibleMessage(out[i].role,
                                         "[tool output pruned to save context]",
                                         out[i].tool_name)
            self.compaction_state.pruned_tool_ids.add(i)
            total -= saved - 8
        return out
