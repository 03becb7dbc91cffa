"""Cursor-proximity context gathering for prediction features.

Rebuild of browser/contextGatheringService.ts (354 LoC): on a cursor move
the service caches a deduplicated list of code snippets around the cursor —
the nearby lines, the enclosing container function, and the definitions of
symbols referenced nearby — which the prediction features consume
(getCachedSnippets).  The reference resolves symbols through VS Code's
language services; this engine analog resolves them lexically: identifiers
near the cursor are matched against definition patterns
(def/class/function/const/assignment) across the workspace's source files.

Constants and snippet hygiene follow the reference: NUM_LINES=3 context
lines, MAX_SNIPPET_LINES=7 cap, comment/blank-line cleaning, multi-newline
normalization, and the visited-interval overlap dedup.
"""

from __future__ import annotations

import os
import re
from typing import Dict, List, Optional, Set, Tuple

NUM_LINES = 3
MAX_SNIPPET_LINES = 7
MAX_DEF_FILES = 50  # workspace scan cap (reference uses open models only)

_SOURCE_EXTS = (".py", ".ts", ".tsx", ".js", ".jsx", ".c", ".h", ".cpp",
                ".hip", ".rs", ".go", ".java", ".rb")

_IDENT_RE = re.compile(r"[A-Za-z_][A-Za-z0-9_]{2,}")
_KEYWORDS = {
    "def", "class", "function", "const", "return", "import", "from", "for",
    "while", "with", "self", "this", "else", "elif", "None", "True", "False",
    "let", "var", "new", "int", "float", "str", "bool", "void", "and", "not",
    "try", "except", "raise", "pass", "lambda", "yield", "async", "await",
}


def _def_pattern(name: str) -> re.Pattern:
    return re.compile(
        r"^\s*(?:def|class|function|const|let|var|struct|enum|interface|type)\s+"
        + re.escape(name) + r"\b"
        + r"|^\s*" + re.escape(name) + r"\s*[:=]\s*", re.M)


class ContextGatheringService:
    def __init__(self, root: str) -> None:
        self.root = os.path.abspath(root)
        self._cache: List[str] = []
        self._files: Dict[str, List[str]] = {}  # uri -> lines (open models)

    # ---- model registry (the reference subscribes to ITextModel adds) ----
    def open_file(self, uri: str, text: Optional[str] = None) -> None:
        if text is None:
            with open(os.path.join(self.root, uri), "r", encoding="utf-8",
                      errors="replace") as f:
                text = f.read()
        self._files[uri] = text.split("\n")

    # ---- snippet hygiene (reference _cleanSnippet/_normalizeSnippet) ----
    @staticmethod
    def _clean(snippet: str) -> str:
        lines = [ln for ln in snippet.split("\n")
                 if ln.strip() and not re.match(r"^/{2,}$", ln.strip())]
        return "\n".join(lines).strip()

    @staticmethod
    def _normalize(snippet: str) -> str:
        return re.sub(r"\n{2,}", "\n", snippet).strip()

    def _snippet_for_range(self, lines: List[str], start: int, end: int) -> str:
        """1-based inclusive range, +-NUM_LINES padding, MAX cap from the end."""
        s = max(start - NUM_LINES, 1)
        e = min(end + NUM_LINES, len(lines))
        if e - s + 1 > MAX_SNIPPET_LINES:
            s = e - MAX_SNIPPET_LINES + 1
        return self._clean("\n".join(lines[s - 1: e]))

    # ---- visited-interval dedup ----
    @staticmethod
    def _visited(uri: str, s: int, e: int,
                 visited: List[Tuple[str, int, int]]) -> bool:
        return any(u == uri and not (e < vs or s > ve)
                   for (u, vs, ve) in visited)

    def _add(self, uri: str, lines: List[str], start: int, end: int,
             snippets: Set[str], visited: List[Tuple[str, int, int]]) -> None:
        if self._visited(uri, start, end, visited):
            return
        visited.append((uri, start, end))
        snip = self._normalize(self._snippet_for_range(lines, start, end))
        if snip:
            snippets.add(snip)

    # ---- lexical symbol resolution ----
    def _symbols_near(self, lines: List[str], line: int) -> List[str]:
        s = max(line - NUM_LINES, 1)
        e = min(line + NUM_LINES, len(lines))
        text = "\n".join(lines[s - 1: e])
        out: List[str] = []
        for m in _IDENT_RE.finditer(text):
            name = m.group(0)
            if name not in _KEYWORDS and name not in out:
                out.append(name)
        return out[:12]

    def _workspace_files(self) -> List[str]:
        found: List[str] = list(self._files)
        for dirpath, dirnames, filenames in os.walk(self.root):
            dirnames[:] = [d for d in dirnames
                           if not d.startswith(".") and d != "node_modules"]
            for fn in filenames:
                if fn.endswith(_SOURCE_EXTS):
                    rel = os.path.relpath(os.path.join(dirpath, fn), self.root)
                    if rel not in found:
                        found.append(rel)
                if len(found) >= MAX_DEF_FILES:
                    return found
        return found

    def _find_definition(self, name: str) -> Optional[Tuple[str, List[str], int]]:
        pat = _def_pattern(name)
        for uri in self._workspace_files():
            if uri in self._files:
                lines = self._files[uri]
            else:
                try:
                    with open(os.path.join(self.root, uri), "r",
                              encoding="utf-8", errors="replace") as f:
                        lines = f.read().split("\n")
                except OSError:
                    continue
            m = pat.search("\n".join(lines))
            if m:
                lineno = "\n".join(lines)[: m.start()].count("\n") + 1
                return uri, lines, lineno
        return None

    def _container_start(self, lines: List[str], line: int) -> Optional[int]:
        """Nearest enclosing def/class/function line above (indent-lexical)."""
        for ln in range(min(line, len(lines)), 0, -1):
            if re.match(r"^\s*(def|class|function|fn)\b", lines[ln - 1]):
                return ln
        return None

    # ---- the public surface (updateCache / getCachedSnippets) ----
    def update_cache(self, uri: str, line: int) -> List[str]:
        if uri not in self._files:
            self.open_file(uri)
        lines = self._files[uri]
        line = max(1, min(line, len(lines)))
        snippets: Set[str] = set()
        visited: List[Tuple[str, int, int]] = []

        # 1. nearby snippet around the cursor
        self._add(uri, lines, line, line, snippets, visited)
        # 2. definitions of symbols near the cursor
        for name in self._symbols_near(lines, line):
            hit = self._find_definition(name)
            if hit:
                duri, dlines, dline = hit
                self._add(duri, dlines, dline, dline, snippets, visited)
        # 3. the enclosing container function and ITS nearby symbols
        cstart = self._container_start(lines, line)
        if cstart is not None:
            self._add(uri, lines, cstart, cstart, snippets, visited)
            for name in self._symbols_near(lines, cstart):
                hit = self._find_definition(name)
                if hit:
                    duri, dlines, dline = hit
                    self._add(duri, dlines, dline, dline, snippets, visited)

        self._cache = sorted(snippets)
        return self._cache

    def get_cached_snippets(self) -> List[str]:
        return list(self._cache)
