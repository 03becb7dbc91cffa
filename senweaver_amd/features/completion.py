"""Autocomplete (FIM) + next-edit prediction services.

Rebuilds the reference's completion surface:
- AutocompleteService (browser/autocompleteService.ts): FIM inline
  completion with an LRU cache keyed on (prefix tail, suffix head) (:72);
- EditPredictionService (browser/editPredictionService.ts) with its
  PredictionCache (:90): predicts the next edit from recent edit history
  plus cursor context.

The backbone serves FIM through the standard sentinel prompt
(<|fim_prefix|> / <|fim_suffix|> / <|fim_middle|> — the local analog of
the reference's per-provider sendFIM implementations).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from ..utils.cache import LRUCache

FIM_PREFIX = "<|fim_prefix|>"
FIM_SUFFIX = "<|fim_suffix|>"
FIM_MIDDLE = "<|fim_middle|>"

CACHE_CAPACITY = 64
PREFIX_CONTEXT_CHARS = 2000
SUFFIX_CONTEXT_CHARS = 500


def build_fim_prompt(prefix: str, suffix: str) -> str:
    return (f"{FIM_PREFIX}{prefix[-PREFIX_CONTEXT_CHARS:]}"
            f"{FIM_SUFFIX}{suffix[:SUFFIX_CONTEXT_CHARS]}{FIM_MIDDLE}")


class AutocompleteService:
    def __init__(self, backend, max_new_tokens: int = 32) -> None:
        self._backend = backend
        self._max_new_tokens = max_new_tokens
        self.cache: LRUCache = LRUCache(CACHE_CAPACITY)
        self.enabled = True

    def _cache_key(self, prefix: str, suffix: str) -> str:
        return prefix[-200:] + "\x00" + suffix[:100]

    def complete(self, prefix: str, suffix: str = "") -> Optional[str]:
        """Synchronous FIM completion (cached)."""
        if not self.enabled:
            return None
        key = self._cache_key(prefix, suffix)
        hit = self.cache.get(key)
        if hit is not None:
            return hit
        prompt = build_fim_prompt(prefix, suffix)
        out = self._backend.generate(prompt, max_new_tokens=self._max_new_tokens)
        # keep the first line only, like inline completion UIs
        out = out.split("\n")[0]
        self.cache.put(key, out)
        return out


@dataclass
class EditEvent:
    uri: str
    before: str
    after: str


class PredictionCache:
    """Keyed on (uri, cursor context) — editPredictionService.ts:90."""

    def __init__(self, capacity: int = 32) -> None:
        self._lru: LRUCache = LRUCache(capacity)

    def get(self, uri: str, context: str):
        return self._lru.get(f"{uri}\x00{context[-160:]}")

    def put(self, uri: str, context: str, prediction: str) -> None:
        self._lru.put(f"{uri}\x00{context[-160:]}", prediction)


class EditPredictionService:
    def __init__(self, backend, max_history: int = 8,
                 context_gatherer=None) -> None:
        self._backend = backend
        self._history: List[EditEvent] = []
        self._max_history = max_history
        self.cache = PredictionCache()
        # the editor-side half (contextGatheringService.ts): cursor-proximity
        # snippets feed the prediction prompt
        self._gatherer = context_gatherer

    def record_edit(self, uri: str, before: str, after: str) -> None:
        self._history.append(EditEvent(uri, before, after))
        if len(self._history) > self._max_history:
            self._history = self._history[-self._max_history:]

    def predict_next_edit(self, uri: str, cursor_context: str) -> str:
        cached = self.cache.get(uri, cursor_context)
        if cached is not None:
            return cached
        history = "\n".join(
            f"- in {e.uri}: {e.before[:80]!r} -> {e.after[:80]!r}"
            for e in self._history[-4:])
        gathered = ""
        if self._gatherer is not None:
            snips = self._gatherer.get_cached_snippets()[:4]
            if snips:
                gathered = "Workspace context:\n" + "\n---\n".join(snips) + "\n\n"
        prompt = (f"{gathered}Recent edits:\n{history}\n\n"
                  f"Code near cursor in {uri}:\n{cursor_context[-500:]}\n\n"
                  "Predict the next edit the user will make (answer with the "
                  "edited code only):")
        pred = self._backend.generate(prompt, max_new_tokens=48)
        self.cache.put(uri, cursor_context, pred)
        return pred
