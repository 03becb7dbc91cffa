"""Marker (lint diagnostic) service + checker.

Rebuild of browser/_markerCheckService.ts on the engine's substrate: the
reference reads VS Code's IMarkerService, filters Error-severity markers and
queries quick-fix code-action providers for each.  Here the marker store is
owned by this module (populated by the built-in linters or any caller), and
"code action providers" are plain callables registered per language.  The
read_lint_errors tool consumes `read()` for its per-file diagnostics.
"""

from __future__ import annotations

import ast
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

# MarkerSeverity values mirror VS Code's (markers.ts)
ERROR = 8
WARNING = 4
INFO = 2
HINT = 1


@dataclass
class Marker:
    resource: str
    severity: int
    message: str
    startLineNumber: int
    startColumn: int = 1
    endLineNumber: int = 0
    endColumn: int = 1

    def __post_init__(self) -> None:
        if not self.endLineNumber:
            self.endLineNumber = self.startLineNumber


@dataclass
class CodeAction:
    title: str
    kind: str = "quickfix"
    edit: Optional[dict] = None  # {"uri", "range", "newText"}


class MarkerService:
    """Per-resource marker store with change listeners (IMarkerService)."""

    def __init__(self) -> None:
        self._markers: Dict[str, List[Marker]] = {}
        self._listeners: List[Callable[[str], None]] = []

    def changed(self, resource: str, markers: List[Marker]) -> None:
        if markers:
            self._markers[resource] = list(markers)
        else:
            self._markers.pop(resource, None)
        for fn in self._listeners:
            fn(resource)

    def read(self, resource: Optional[str] = None,
             severity: Optional[int] = None) -> List[Marker]:
        out: List[Marker] = []
        for res, ms in self._markers.items():
            if resource is not None and res != resource:
                continue
            out.extend(m for m in ms
                       if severity is None or m.severity == severity)
        return out

    def on_marker_changed(self, fn: Callable[[str], None]) -> None:
        self._listeners.append(fn)


def python_lint(resource: str, text: str) -> List[Marker]:
    """Built-in provider: Python syntax errors via ast.parse."""
    try:
        ast.parse(text, filename=resource)
        return []
    except SyntaxError as e:
        return [Marker(resource, ERROR, e.msg or "syntax error",
                       e.lineno or 1, (e.offset or 1))]


class MarkerCheckService:
    """The checker: collects Error markers and queries quick-fix providers
    (the reference's provideCodeActions(... only='quickfix') loop)."""

    def __init__(self, marker_service: MarkerService) -> None:
        self._markers = marker_service
        self._providers: Dict[str, List[Callable[[Marker, str], List[CodeAction]]]] = {}

    def register_code_action_provider(
            self, language: str,
            provider: Callable[[Marker, str], List[CodeAction]]) -> None:
        self._providers.setdefault(language, []).append(provider)

    def check(self, get_text: Callable[[str], str],
              language_of: Callable[[str], str]) -> List[dict]:
        """One pass over all Error markers -> [{marker, actions}]."""
        results = []
        for marker in self._markers.read(severity=ERROR):
            actions: List[CodeAction] = []
            lang = language_of(marker.resource)
            for provider in self._providers.get(lang, []):
                try:
                    actions.extend(provider(marker, get_text(marker.resource)))
                except Exception:
                    continue  # a broken provider never breaks the check
            results.append({"marker": marker, "actions": actions})
        return results
