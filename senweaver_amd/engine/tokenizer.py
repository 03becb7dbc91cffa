"""Deterministic offline tokenizer.

There is no network access for real tokenizer vocabularies, so this is a
stable hash tokenizer: whitespace/punctuation word pieces map to fixed ids in
[256, vocab) via BLAKE2 (deterministic across processes and ranks — critical
for the RCCL candidate-parallel scorer, where every rank must tokenize a
candidate identically); raw bytes 0-255 are reserved for byte fallback.
Token ids — not text round-tripping — are what scoring and the synthetic
benches need; detokenization emits placeholder word forms.
"""

from __future__ import annotations

import hashlib
import re
from typing import List

_WORD_RE = re.compile(r"\w+|[^\w\s]|\s")

BOS = 1
EOS = 2
ROLE_USER = 3
ROLE_ASSISTANT = 4
ROLE_TOOL = 5
ROLE_SYSTEM = 6
_SPECIAL_MAX = 16


class HashTokenizer:
    def __init__(self, vocab_size: int) -> None:
        self.vocab_size = vocab_size

    def _piece_id(self, piece: str) -> int:
        if len(piece) == 1 and ord(piece) < 128:
            return 16 + ord(piece)  # stable ASCII band above specials
        h = int.from_bytes(hashlib.blake2b(piece.encode(), digest_size=8).digest(), "big")
        span = self.vocab_size - 256
        return 256 + (h % span)

    def encode(self, text: str, max_tokens: int | None = None) -> List[int]:
        out: List[int] = []
        for piece in _WORD_RE.findall(text):
            if piece.isspace():
                continue
            out.append(self._piece_id(piece))
            if max_tokens is not None and len(out) >= max_tokens:
                break
        return out

    def decode(self, ids: List[int]) -> str:
        return " ".join(f"tok{i}" for i in ids)
