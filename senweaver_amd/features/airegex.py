"""AIRegexService — natural-language to regex (reference aiRegexService.ts).

The reference asks the configured LLM to turn a plain-language description
into a search regex, validates it, and falls back to a literal-escaped
pattern when generation fails.  Here the generator is the local backbone
(any object with ``generate(prompt, max_new_tokens) -> str``); the
validation, extraction, and fallback semantics are the service contract.
"""

from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Optional

_PROMPT = (
    "Convert this description of a text pattern into a single regular "
    "expression. Return ONLY the regex, no explanation, no delimiters.\n"
    "Description: {description}\nRegex:"
)


@dataclass
class AIRegexResult:
    pattern: str
    is_fallback: bool
    error: Optional[str] = None


class AIRegexService:
    def __init__(self, backend=None) -> None:
        self._backend = backend  # None => always literal fallback (offline)

    def generate(self, description: str) -> AIRegexResult:
        """LLM regex with validation; literal-escape fallback on any failure."""
        if self._backend is not None:
            try:
                raw = self._backend.generate(_PROMPT.format(description=description),
                                             max_new_tokens=64)
                pattern = self.extract_regex(raw)
                if pattern:
                    re.compile(pattern)
                    return AIRegexResult(pattern=pattern, is_fallback=False)
            except re.error as e:
                return AIRegexResult(pattern=re.escape(description),
                                     is_fallback=True, error=f"invalid regex: {e}")
            except Exception as e:  # generation failure
                return AIRegexResult(pattern=re.escape(description),
                                     is_fallback=True, error=str(e))
        return AIRegexResult(pattern=re.escape(description), is_fallback=True)

    @staticmethod
    def extract_regex(raw: str) -> str:
        """First non-empty line, stripped of code fences / slash delimiters."""
        for line in raw.splitlines():
            s = line.strip()
            if not s or s.startswith("```"):
                continue
            if len(s) > 2 and s.startswith("/") and s.rstrip("gimsuxy").endswith("/"):
                s = s.rstrip("gimsuxy")[1:-1]
            return s
        return ""
