"""Python client for the native daemon (the renderer-proxy analog).

Mirrors sendLLMMessageService.ts: request-id-keyed hook registry; send
returns the requestId which is the abort token; events dispatched to the
registered callbacks.
"""

from __future__ import annotations

import json
import os
import socket
import subprocess
import threading
import time
import uuid
from typing import Callable, Dict, List, Optional

_HERE = os.path.dirname(os.path.abspath(__file__))
DAEMON_BIN = os.path.join(_HERE, "_bin", "senweaver_daemon")


def build_daemon(force: bool = False) -> str:
    """Compile the C++ daemon with g++ (no GPU/torch dependency)."""
    src = os.path.join(_HERE, "daemon.cpp")
    os.makedirs(os.path.dirname(DAEMON_BIN), exist_ok=True)
    if force or (not os.path.exists(DAEMON_BIN)
                 or os.path.getmtime(DAEMON_BIN) < os.path.getmtime(src)):
        subprocess.run(["g++", "-O2", "-std=c++17", "-o", DAEMON_BIN, src], check=True)
    return DAEMON_BIN


class DaemonClient:
    def __init__(self, socket_path: str) -> None:
        self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self._sock.connect(socket_path)
        self._hooks: Dict[str, Dict[str, Callable]] = {}
        self._lock = threading.Lock()
        self._closed = False
        # debug ring of recent events (bounded — a long-lived client must
        # not grow without bound); control-plane replies use _waiters
        self.events: List[dict] = []
        self._waiters: Dict[str, list] = {}  # event name -> [Event, payload]
        self._reader = threading.Thread(target=self._read_loop, daemon=True)
        self._reader.start()

    def _read_loop(self) -> None:
        buf = b""
        while not self._closed:
            try:
                chunk = self._sock.recv(65536)
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            while b"\n" in buf:
                line, buf = buf.split(b"\n", 1)
                if not line:
                    continue
                try:
                    msg = json.loads(line)
                except ValueError:
                    continue
                ev = msg.get("event", "")
                rid = msg.get("requestId", "")
                with self._lock:
                    self.events.append(msg)
                    if len(self.events) > 2048:
                        del self.events[:1024]
                    w = self._waiters.pop(ev, None)
                    hooks = self._hooks.get(rid, {})
                    if ev in ("onFinalMessage", "onError", "onAbort"):
                        self._hooks.pop(rid, None)  # terminal: free the entry
                if w is not None:
                    w[1] = msg
                    w[0].set()
                fn = hooks.get(ev)
                if fn:
                    try:
                        fn(msg)
                    except Exception:
                        pass

    def _send(self, obj: dict) -> None:
        self._sock.sendall((json.dumps(obj) + "\n").encode())

    def send_llm_message(self, messages: List[dict], on_text=None, on_final=None,
                         on_error=None, chat_mode: Optional[str] = None,
                         max_new_tokens: int = 64,
                         model_options: Optional[dict] = None,
                         raw_prompt: Optional[str] = None) -> str:
        rid = str(uuid.uuid4())
        with self._lock:
            self._hooks[rid] = {"onText": on_text or (lambda m: None),
                                "onFinalMessage": on_final or (lambda m: None),
                                "onError": on_error or (lambda m: None),
                                "onAbort": lambda m: None}
        req = {"method": "sendLLMMessage", "requestId": rid,
               "messages": messages, "chatMode": chat_mode,
               "maxNewTokens": max_new_tokens}
        if model_options:
            req["modelOptions"] = model_options
        if raw_prompt is not None:
            req["rawPrompt"] = raw_prompt
        self._send(req)
        return rid

    def abort(self, request_id: str) -> None:
        self._send({"method": "abort", "requestId": request_id})

    def _request_reply(self, method: str, event: str, timeout: float):
        """One-shot control-plane roundtrip (pong/listResult/statsResult)."""
        w = [threading.Event(), None]
        with self._lock:
            self._waiters[event] = w
        self._send({"method": method})
        if w[0].wait(timeout=timeout):
            return w[1]
        with self._lock:
            self._waiters.pop(event, None)
        return None

    def ping(self, timeout: float = 30.0) -> bool:
        return self._request_reply("ping", "pong", timeout) is not None

    def stats(self, timeout: float = 30.0):
        """Engine token-usage stats (worker "stats" RPC), or None."""
        msg = self._request_reply("stats", "statsResult", timeout)
        return msg.get("usage") if msg else None

    def shutdown(self) -> None:
        """Ask the daemon to shut down (worker + listener)."""
        self._send({"method": "shutdown"})

    def close(self) -> None:
        self._closed = True
        try:
            self._sock.close()
        except OSError:
            pass

class DaemonBackend:
    """Backend adapter: the agent loop's LLMMessageService speaks to the
    NATIVE daemon instead of an in-process model — the reference's actual
    topology (renderer chat service -> main-process LLM channel).

    stream_generate matches LlamaBackend's contract (cumulative on_chunk,
    should_stop polled between events, abort forwarded to the worker).
    """

    def __init__(self, socket_path: str) -> None:
        self._client = DaemonClient(socket_path)
        if not self._client.ping(timeout=180):
            raise RuntimeError("daemon engine did not answer ping")
        self.model_names = self._fetch_models()

    def _fetch_models(self, timeout: float = 30.0) -> list:
        msg = self._client._request_reply("list", "listResult", timeout)
        return msg.get("models", []) if msg else []

    def stream_generate(self, prompt: str, max_new_tokens: int, should_stop,
                        on_chunk, temperature: float = 0.0, top_p: float = 1.0,
                        sample_seed=None, stop=None) -> str:
        done = threading.Event()
        state = {"text": "", "aborted": False}

        def on_text(m):
            state["text"] = m.get("fullText", "")
            on_chunk(state["text"])
            if should_stop() and not state["aborted"]:
                state["aborted"] = True
                self._client.abort(rid)
                done.set()

        def fin(m):
            state["text"] = m.get("fullText", state["text"])
            done.set()

        opts = None
        if temperature > 0 or stop:
            opts = {"temperature": temperature, "topP": top_p,
                    "sampleSeed": sample_seed, "stop": stop}
        rid = self._client.send_llm_message(
            [], on_text=on_text, on_final=fin,
            on_error=lambda m: done.set(),
            max_new_tokens=max_new_tokens, model_options=opts,
            raw_prompt=prompt)
        done.wait(timeout=900)
        return state["text"]

    def close(self) -> None:
        self._client.close()
