"""JSON helpers that reproduce JavaScript ``JSON.stringify`` behavior.

The reference persists every RL artifact as a ``JSON.stringify`` string under
VS Code IStorageService keys (reference: common/traceCollectorService.ts:330-346,
common/apoService.ts:374-416).  For byte-compatible round-tripping we need:

- ``undefined`` object properties omitted (we model them as ``None`` -> omit,
  via the OMIT sentinel / ``drop_none`` flag per schema field);
- integral floats printed without a trailing ``.0`` (JS has one number type);
- no ASCII escaping of non-ASCII text;
- compact separators (no spaces) for ``JSON.stringify(x)`` and 2-space indent
  for ``JSON.stringify(x, null, 2)``.
"""

from __future__ import annotations

import json
import math
from typing import Any


def _normalize(value: Any) -> Any:
    """Recursively convert values so json.dumps output matches JSON.stringify."""
    if isinstance(value, bool) or value is None:
        return value
    if isinstance(value, float):
        if math.isnan(value) or math.isinf(value):
            # JSON.stringify(NaN/Infinity) emits null
            return None
        if value.is_integer() and abs(value) < 2**53:
            return int(value)
        return value
    if isinstance(value, dict):
        return {k: _normalize(v) for k, v in value.items()}
    if isinstance(value, (list, tuple)):
        return [_normalize(v) for v in value]
    return value


def js_stringify(value: Any, indent: int | None = None) -> str:
    """Equivalent of JSON.stringify(value) / JSON.stringify(value, null, indent)."""
    norm = _normalize(value)
    if indent is None:
        return json.dumps(norm, ensure_ascii=False, separators=(",", ":"))
    return json.dumps(norm, ensure_ascii=False, indent=indent)


def js_parse(text: str) -> Any:
    """Equivalent of JSON.parse."""
    return json.loads(text)


def to_fixed(value: float, digits: int) -> str:
    """JavaScript Number.prototype.toFixed for the ranges used here.

    JS toFixed rounds to nearest, ties resolved by the binary representation;
    Python's format() does the same (both go through the exact double value),
    so a plain format matches for every representable double.
    """
    return f"{value:.{digits}f}"
