"""Code extraction from LLM output + language helpers.

Rebuild of the reference's common/helpers/extractCodeFromResult.ts (the
SurroundingsRemover code-fence stripper, the FIM mid-tag extractor, and the
STREAMING search/replace-block parser whose state machine guarantees that
feeding more text never shrinks the block list or regresses a block's
state) and common/helpers/languageHelpers.ts (markdown-language and
filename -> language id mapping).  Used wherever LLM output becomes code:
apply-edit flows, FIM completion, and the edit_file tool's streamed blocks.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List, Optional, Tuple

ORIGINAL = "<<<<<<< ORIGINAL"
DIVIDER = "======="
FINAL = ">>>>>>> UPDATED"


class SurroundingsRemover:
    """View over s[i..j] with prefix/suffix stripping (streaming-safe:
    removeSuffix also strips a PARTIAL suffix that may still be growing)."""

    def __init__(self, s: str) -> None:
        self.s = s
        self.i = 0
        self.j = len(s) - 1

    def value(self) -> str:
        return self.s[self.i: self.j + 1]

    def remove_prefix(self, prefix: str) -> bool:
        off = 0
        while self.i <= self.j and off < len(prefix):
            if self.s[self.i] != prefix[off]:
                break
            off += 1
            self.i += 1
        return off == len(prefix)

    def remove_suffix(self, suffix: str) -> bool:
        s = self.value()
        for ln in range(min(len(s), len(suffix)), 0, -1):
            if s.endswith(suffix[:ln]):
                self.j -= ln
                return ln == len(suffix)
        return False

    def remove_from_start_until(self, until: str, also_remove_until: bool) -> bool:
        idx = self.s.find(until, self.i)
        if idx == -1:
            return False
        self.i = idx + len(until) if also_remove_until else idx
        return True

    def remove_code_block(self) -> bool:
        if not self.remove_prefix("```"):
            return False
        self.remove_from_start_until("\n", True)  # language line
        j = self.j
        found_end = self.remove_suffix("```")
        if self.j == j:
            found_end = self.remove_suffix("```\n")
        if not found_end:
            return False
        self.remove_suffix("\n")
        return True

    def delta_info(self, recently_added_len: int) -> Tuple[str, str]:
        added_idx = len(self.s) - recently_added_len
        delta = self.s[max(self.i, added_idx): self.j + 1]
        ignored = self.s[max(self.j + 1, added_idx):]
        return delta, ignored


def extract_code_from_regular(text: str, recently_added_len: int = 0
                              ) -> Tuple[str, str, str]:
    pm = SurroundingsRemover(text)
    pm.remove_code_block()
    s = pm.value()
    delta, ignored = pm.delta_info(recently_added_len)
    return s, delta, ignored


def extract_code_from_fim(text: str, mid_tag: str,
                          recently_added_len: int = 0) -> Tuple[str, str, str]:
    pm = SurroundingsRemover(text)
    pm.remove_code_block()
    if pm.remove_prefix(f"<{mid_tag}>"):
        pm.remove_suffix("\n")
        pm.remove_suffix(f"</{mid_tag}>")
    s = pm.value()
    delta, ignored = pm.delta_info(recently_added_len)
    return s, delta, ignored


def ends_with_any_prefix_of(s: str, any_prefix: str) -> Optional[str]:
    for i in range(len(any_prefix), 0, -1):
        if s.endswith(any_prefix[:i]):
            return any_prefix[:i]
    return None


@dataclass
class SearchReplaceBlock:
    state: str  # 'writingOriginal' | 'writingFinal' | 'done'
    orig: str
    final: str


def _substr(s: str, start: int, end: int) -> str:
    return "" if end < start else s[start:end]


def extract_search_replace_blocks(s: str) -> List[SearchReplaceBlock]:
    """Streaming parser: as text grows, the block list never shrinks and a
    block's state only progresses (writingOriginal -> writingFinal -> done)."""
    blocks: List[SearchReplaceBlock] = []
    i = 0
    while True:
        orig_marker = s.find(ORIGINAL, i)
        if orig_marker == -1:
            return blocks
        nl_after_orig = s.find("\n", orig_marker)
        if nl_after_orig == -1:
            return blocks  # incomplete stream
        if s[orig_marker + len(ORIGINAL): nl_after_orig].strip():
            i = orig_marker + 1
            continue
        orig_start = nl_after_orig + 1
        i = orig_start

        div_marker = s.find("\n" + DIVIDER, i)
        if div_marker == -1:
            part = ends_with_any_prefix_of(s, "\n" + DIVIDER)
            cut = len(part) if part else 0
            blocks.append(SearchReplaceBlock(
                "writingOriginal", _substr(s, orig_start, len(s) - cut), ""))
            return blocks
        nl_after_div = s.find("\n", div_marker + 1)
        if nl_after_div == -1:
            blocks.append(SearchReplaceBlock(
                "writingOriginal", _substr(s, orig_start, div_marker), ""))
            return blocks

        orig_done = _substr(s, orig_start, div_marker)
        div_start = nl_after_div + 1
        i = div_start

        final_raw = s.find(FINAL, i)
        if final_raw == -1:
            part = (ends_with_any_prefix_of(s, "\n" + FINAL)
                    or ends_with_any_prefix_of(s, FINAL))
            cut = len(part) if part else 0
            blocks.append(SearchReplaceBlock(
                "writingFinal", orig_done, _substr(s, div_start, len(s) - cut)))
            return blocks
        final_marker = final_raw
        preceded_by_nl = final_raw > 0 and s[final_raw - 1] == "\n"
        if preceded_by_nl:
            final_marker = final_raw - 1
        if final_marker < i:
            i = final_raw + 1
            continue
        final_done = _substr(s, div_start,
                             final_marker if preceded_by_nl else final_raw)
        nl_after_final = s.find("\n", final_raw + len(FINAL))
        if nl_after_final == -1:
            nl_after_final = len(s)
        i = nl_after_final + 1
        blocks.append(SearchReplaceBlock("done", orig_done, final_done))


# ---------------------------------------------------------------------------
# Language helpers (languageHelpers.ts)
# ---------------------------------------------------------------------------

_EXT_TO_LANG = {
    ".py": "python", ".js": "javascript", ".jsx": "javascriptreact",
    ".ts": "typescript", ".tsx": "typescriptreact", ".c": "c", ".h": "c",
    ".cpp": "cpp", ".cc": "cpp", ".hpp": "cpp", ".hip": "cpp", ".cu": "cuda",
    ".rs": "rust", ".go": "go", ".java": "java", ".kt": "kotlin",
    ".rb": "ruby", ".php": "php", ".cs": "csharp", ".swift": "swift",
    ".sh": "shellscript", ".bash": "shellscript", ".zsh": "shellscript",
    ".md": "markdown", ".json": "json", ".yaml": "yaml", ".yml": "yaml",
    ".toml": "toml", ".xml": "xml", ".html": "html", ".htm": "html",
    ".css": "css", ".scss": "scss", ".sql": "sql", ".r": "r", ".lua": "lua",
    ".pl": "perl", ".hs": "haskell", ".scala": "scala", ".dart": "dart",
    ".vue": "vue", ".svelte": "svelte", ".txt": "plaintext",
}

_MD_ALIASES = {
    "js": "javascript", "ts": "typescript", "py": "python", "rb": "ruby",
    "sh": "shellscript", "bash": "shellscript", "shell": "shellscript",
    "zsh": "shellscript", "c++": "cpp", "c#": "csharp", "cs": "csharp",
    "yml": "yaml", "golang": "go", "rs": "rust", "kt": "kotlin",
    "plain": "plaintext", "text": "plaintext", "": "plaintext",
}


def filename_to_language(path: str) -> str:
    """Language id from a filename (the detectLanguage uri branch)."""
    name = os.path.basename(path).lower()
    if name == "dockerfile":
        return "dockerfile"
    if name == "makefile":
        return "makefile"
    ext = os.path.splitext(name)[1]
    return _EXT_TO_LANG.get(ext, "plaintext")


def markdown_language_to_id(lang: str) -> str:
    """Normalize a fenced-code-block language tag (convertToVscodeLang)."""
    lang = (lang or "").strip().lower()
    if lang in _MD_ALIASES:
        return _MD_ALIASES[lang]
    if lang in _EXT_TO_LANG.values():
        return lang
    return _MD_ALIASES.get(lang, lang or "plaintext")
