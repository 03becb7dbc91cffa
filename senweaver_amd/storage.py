"""Key/value storage with the semantics of VS Code's IStorageService.

The reference stores every RL artifact as a JSON *string* under a flat key
(APPLICATION scope, MACHINE target — reference common/traceCollectorService.ts:215-217,
common/apoService.ts:273-275).  This module provides the same get/store API over a
single JSON file on disk, so trace/segment/beam-state data exported from the
reference re-imports cleanly and vice versa.

Storage keys (identical to the reference):
  senweaver.traceCollector.data / .feedbacks / .uploadConfig / .uploadedIds
  senweaver.apo.data / .config / .segments / .beamState / .gradients
"""

from __future__ import annotations

import json
import os
import tempfile
import threading
from typing import Dict, Optional


class MemoryStorage:
    """In-memory storage (test double, like VS Code's in-memory storage service)."""

    def __init__(self) -> None:
        self._data: Dict[str, str] = {}
        self._lock = threading.Lock()

    def get(self, key: str, default: Optional[str] = None) -> Optional[str]:
        with self._lock:
            return self._data.get(key, default)

    def store(self, key: str, value: str) -> None:
        with self._lock:
            self._data[key] = value

    def remove(self, key: str) -> None:
        with self._lock:
            self._data.pop(key, None)

    def keys(self):
        with self._lock:
            return list(self._data.keys())


class FileStorage(MemoryStorage):
    """Durable storage: one JSON file mapping key -> JSON-string value.

    Writes are atomic (tmp file + rename) so a crash mid-flush never corrupts
    RL state — the reference's crash-safe-resume contract (README.md:143 of the
    reference: all RL data persists, restart loses nothing).
    """

    def __init__(self, path: str) -> None:
        super().__init__()
        self._path = path
        self._load()

    def _load(self) -> None:
        try:
            with open(self._path, "r", encoding="utf-8") as f:
                data = json.load(f)
            if isinstance(data, dict):
                self._data.update({k: v for k, v in data.items() if isinstance(v, str)})
        except (OSError, ValueError):
            pass  # silent, like the reference's storage-load failure path

    def flush(self) -> None:
        with self._lock:
            snapshot = dict(self._data)
        d = os.path.dirname(os.path.abspath(self._path)) or "."
        os.makedirs(d, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=d, prefix=".storage-", suffix=".tmp")
        try:
            with os.fdopen(fd, "w", encoding="utf-8") as f:
                json.dump(snapshot, f, ensure_ascii=False)
            os.replace(tmp, self._path)
        except OSError:
            try:
                os.unlink(tmp)
            except OSError:
                pass

    def store(self, key: str, value: str) -> None:
        super().store(key, value)

    def remove(self, key: str) -> None:
        super().remove(key)


# Storage key constants (must match the reference byte-for-byte)
TRACE_STORAGE_KEY = "senweaver.traceCollector.data"
TRACE_FEEDBACK_KEY = "senweaver.traceCollector.feedbacks"
TRACE_UPLOAD_CONFIG_KEY = "senweaver.traceCollector.uploadConfig"
TRACE_UPLOADED_IDS_KEY = "senweaver.traceCollector.uploadedIds"
APO_STORAGE_KEY = "senweaver.apo.data"
APO_CONFIG_KEY = "senweaver.apo.config"
APO_SEGMENTS_KEY = "senweaver.apo.segments"
APO_BEAM_STATE_KEY = "senweaver.apo.beamState"
APO_GRADIENTS_KEY = "senweaver.apo.gradients"
