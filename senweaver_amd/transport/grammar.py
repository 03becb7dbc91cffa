"""Streaming grammar parsers over the token stream.

Rebuilds the reference's extractGrammar layer (electron-main/llmMessage/
extractGrammar.ts): the reasoning wrapper splits ``<think>...</think>`` out
of the cumulative text stream with partial-tag buffering at chunk
boundaries (:17-139), and the XML tool wrapper incrementally detects
``<tool_name>...`` calls, parsing partially-streamed params so the UI/agent
loop can act before the call is complete (:324-419).

Contract (sendLLMMessage.impl.ts:400-452): ``on_text`` receives CUMULATIVE
fullText/fullReasoning strings, never deltas — consumers recompute
``new_text = full_text[len(prev):]``.
"""

from __future__ import annotations

import re
import uuid
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple


OnText = Callable[..., None]  # kwargs: full_text, full_reasoning, tool_call
OnFinal = Callable[..., None]


def ends_with_any_prefix_of(s: str, any_prefix: str) -> Optional[str]:
    """Longest non-empty prefix of ``any_prefix`` that ``s`` ends with."""
    for i in range(len(any_prefix), 0, -1):
        if s.endswith(any_prefix[:i]):
            return any_prefix[:i]
    return None


# Parameter aliases for model compatibility (extractGrammar.ts:278-309);
# ambiguous aliases ('file', 'folder', 'content', ...) deliberately absent.
PARAM_ALIASES: Dict[str, str] = {
    "path": "uri", "file_path": "uri", "filepath": "uri", "directory": "uri",
    "dir": "uri", "target": "uri", "location": "uri",
    "file_content": "new_content",
    "search": "query", "search_query": "query", "keyword": "query",
    "keywords": "query", "term": "query",
    "blocks": "search_replace_blocks", "changes": "search_replace_blocks",
    "edits": "search_replace_blocks", "replacements": "search_replace_blocks",
    "recursive": "is_recursive", "isRecursive": "is_recursive",
    "regex": "is_regex", "isRegex": "is_regex", "use_regex": "is_regex",
}


@dataclass
class RawToolCall:
    name: str
    raw_params: Dict[str, str] = field(default_factory=dict)
    done_params: List[str] = field(default_factory=list)
    is_done: bool = False
    id: str = ""


def trim_before_and_after_newlines(s: str) -> str:
    """Strip whitespace-only head up to the first newline and whitespace-only
    tail after the last newline (extractGrammar.ts:424-441)."""
    if not s:
        return s
    first = s.find("\n")
    if first != -1 and s[:first].strip() == "":
        s = s[first + 1:]
    last = s.rfind("\n")
    if last != -1 and s[last + 1:].strip() == "":
        s = s[:last]
    return s


def parse_xml_prefix_to_tool_call(tool_name: str, tool_id: str, s: str,
                                  params_of_tool: Dict[str, List[str]]) -> RawToolCall:
    """Parse a (possibly partial) ``<tool>...`` XML prefix into a RawToolCall."""
    call = RawToolCall(name=tool_name, id=tool_id)
    open_tag = f"<{tool_name}>"
    i = s.find(open_tag)
    if i == -1:
        return call
    j = s.rfind(f"</{tool_name}>")
    if j == -1:
        j = len(s)
    else:
        call.is_done = True
    body = s[i + len(open_tag): j]

    allowed = params_of_tool.get(tool_name) or []
    if not allowed:
        return call
    name_to_std = {p: p for p in allowed}
    for alias, std in PARAM_ALIASES.items():
        if std in allowed:
            name_to_std[alias] = std

    pos = 0
    latest: Optional[str] = None
    for _ in range(10):  # bounded like the reference
        matched_std = matched_tag = None
        for possible in name_to_std:
            idx = body.find(f"<{possible}>", pos)
            if idx != -1:
                pos = idx + len(possible) + 2
                matched_std, matched_tag = name_to_std[possible], possible
                break
        if matched_std is None:
            if latest is not None:
                call.raw_params[latest] = call.raw_params.get(latest, "") + body[pos:]
            break
        latest = matched_std
        call.raw_params[latest] = ""
        # close tag: same tag name first, then any
        close_found = False
        tags = [matched_tag] + [p for p in name_to_std if p != matched_tag]
        for possible in tags:
            close = f"</{possible}>"
            idx = body.find(close, pos)
            if idx != -1:
                call.raw_params[latest] += body[pos:idx]
                pos = idx + len(close)
                close_found = True
                break
        if not close_found:
            call.raw_params[latest] += body[pos:]
            break
        call.done_params.append(latest)
    for p in list(call.raw_params):
        call.raw_params[p] = trim_before_and_after_newlines(call.raw_params[p])
    return call


class ReasoningExtractor:
    """Streaming <think>...</think> splitter (cumulative-string contract)."""

    def __init__(self, think_tags: Tuple[str, str] = ("<think>", "</think>")) -> None:
        if not think_tags[0] or not think_tags[1]:
            raise ValueError("think tags must be non-empty")
        self.tags = think_tags
        self.full_text = ""
        self.full_reasoning = ""
        self._latest_idx = 0
        self._found1 = False
        self._found2 = False

    def feed(self, cumulative_text: str) -> Tuple[str, str, bool]:
        """Feed the cumulative raw text; returns (text, reasoning, emit).

        Single consumed-index state machine: ``_latest_idx`` is the position
        in the cumulative string that has been classified so far, and each
        call loops through as many state transitions (text -> reasoning ->
        text) as the string allows — so a one-shot feed of the complete text
        splits fully, and arbitrary chunk boundaries give identical results
        (chunking-invariance property test in tests/test_grammar.py).  A
        trailing partial open/close tag is held back until more text (or
        finalize) resolves it.
        """
        t1, t2 = self.tags
        s = cumulative_text
        emitted = False
        while True:
            if not self._found1:
                idx = s.find(t1, self._latest_idx)
                if idx != -1:
                    if idx > self._latest_idx:
                        self.full_text += s[self._latest_idx: idx]
                        emitted = True
                    self._latest_idx = idx + len(t1)
                    self._found1 = True
                    continue
                keep = len(s)
                p = ends_with_any_prefix_of(s, t1)
                if p:
                    keep = len(s) - len(p)
                if keep > self._latest_idx:
                    self.full_text += s[self._latest_idx: keep]
                    self._latest_idx = keep
                    emitted = True
                break
            if not self._found2:
                idx = s.find(t2, self._latest_idx)
                if idx != -1:
                    if idx > self._latest_idx:
                        self.full_reasoning += s[self._latest_idx: idx]
                        emitted = True
                    self._latest_idx = idx + len(t2)
                    self._found2 = True
                    continue
                keep = len(s)
                p = ends_with_any_prefix_of(s, t2)
                if p:
                    keep = len(s) - len(p)
                if keep > self._latest_idx:
                    self.full_reasoning += s[self._latest_idx: keep]
                    self._latest_idx = keep
                    emitted = True
                break
            if len(s) > self._latest_idx:
                self.full_text += s[self._latest_idx:]
                self._latest_idx = len(s)
                emitted = True
            break
        return self.full_text, self.full_reasoning, emitted

    def finalize(self, cumulative_text: str) -> Tuple[str, str]:
        """Final (text, reasoning) after the stream ends (extractGrammar :117-127):
        the accumulated split already excludes the think tags themselves.
        A trailing partial tag prefix is flushed as literal content of the
        active region (the stream is over — it can never complete)."""
        self.feed(cumulative_text)
        s = cumulative_text
        if self._latest_idx < len(s):
            tail = s[self._latest_idx:]
            if self._found1 and not self._found2:
                self.full_reasoning += tail
            else:
                self.full_text += tail
            self._latest_idx = len(s)
        return self.full_text, self.full_reasoning


class XMLToolExtractor:
    """Incremental XML tool-call detector over the cumulative text stream."""

    def __init__(self, chat_mode: Optional[str], extra_tools: Optional[List[dict]] = None) -> None:
        from ..tools.registry import available_tools
        tools = available_tools(chat_mode, extra_tools) if chat_mode else None
        self.enabled = bool(tools)
        self.params_of_tool = {t["name"]: list(t.get("params", [])) for t in (tools or [])}
        self.open_tags = [f"<{n}>" for n in self.params_of_tool]
        self.tool_id = str(uuid.uuid4())
        self.full_text = ""
        self._true_full = ""
        self._open_tag_buffer = ""
        self._prev_len = 0
        self._found_open: Optional[Tuple[int, str]] = None
        self.latest_tool_call: Optional[RawToolCall] = None

    def feed(self, cumulative_text: str) -> Tuple[str, Optional[RawToolCall]]:
        if not self.enabled:
            self.full_text = cumulative_text
            return self.full_text, None
        new_text = cumulative_text[self._prev_len:]
        self._prev_len = len(cumulative_text)
        self._true_full = cumulative_text

        if self._found_open is None:
            buffered = self._open_tag_buffer + new_text
            partial = any(ends_with_any_prefix_of(buffered, t) for t in self.open_tags)
            if partial:
                self._open_tag_buffer += new_text
            else:
                self.full_text += self._open_tag_buffer
                self._open_tag_buffer = ""
                self.full_text += new_text
                for tag in self.open_tags:
                    idx = self.full_text.find(tag)
                    if idx != -1:
                        self._found_open = (idx, tag[1:-1])
                        self.full_text = self.full_text[:idx]
                        break
        if self._found_open is not None:
            idx, name = self._found_open
            self.latest_tool_call = parse_xml_prefix_to_tool_call(
                name, self.tool_id, self._true_full[idx:], self.params_of_tool)
        return self.full_text, self.latest_tool_call

    def finalize(self, cumulative_text: str) -> Tuple[str, Optional[RawToolCall]]:
        self.feed(cumulative_text)
        self.full_text = self.full_text.rstrip()
        return self.full_text, self.latest_tool_call
