"""Upload sinks for the TraceCollector.

The reference posts the v2.0.0 incremental payload to
``{apiBaseUrl}/api/traces`` over HTTPS (traceCollectorService.ts:797-899).
There is no network in this environment, so these are the engine's local
transports with the same contract — a sink receives one payload dict and
returns True when it has durably accepted it (only then does the collector
advance its uploadedIds watermark):

  - ``FileSink``: appends each payload as one JSON line to a local file
    (fsync'd) — the durable-archive transport.
  - ``UDSSink``: sends the payload over a Unix domain socket as one
    newline-delimited JSON message and requires a ``{"ok": true}`` ack —
    the daemon/collector-service transport, mirroring the request/response
    of the reference's HTTP POST.
"""

from __future__ import annotations

import json
import os
import socket
from typing import Any, Dict

from ..utils.jsonutil import js_stringify


class FileSink:
    def __init__(self, path: str) -> None:
        self.path = path

    def __call__(self, payload: Dict[str, Any]) -> bool:
        line = js_stringify(payload)
        os.makedirs(os.path.dirname(os.path.abspath(self.path)), exist_ok=True)
        with open(self.path, "a", encoding="utf-8") as f:
            f.write(line + "\n")
            f.flush()
            os.fsync(f.fileno())
        return True


class UDSSink:
    def __init__(self, socket_path: str, timeout: float = 10.0) -> None:
        self.socket_path = socket_path
        self.timeout = timeout

    def __call__(self, payload: Dict[str, Any]) -> bool:
        msg = (js_stringify(payload) + "\n").encode("utf-8")
        with socket.socket(socket.AF_UNIX, socket.SOCK_STREAM) as s:
            s.settimeout(self.timeout)
            s.connect(self.socket_path)
            s.sendall(msg)
            buf = b""
            while not buf.endswith(b"\n"):
                chunk = s.recv(4096)
                if not chunk:
                    break
                buf += chunk
        try:
            ack = json.loads(buf.decode("utf-8"))
        except ValueError:
            return False
        return bool(ack.get("ok"))
