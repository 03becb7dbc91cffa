"""MCP (Model Context Protocol) client over stdio JSON-RPC.

Capability analog of the reference's MCPService + mcpChannel
(common/mcpService.ts, electron-main/mcpChannel.ts): reads an ``mcp.json``
config, launches each server as a subprocess, performs the MCP
initialize/tools-list handshake (JSON-RPC 2.0 over newline-delimited stdio)
and exposes discovered tools to the agent loop (they carry the 'MCP tools'
approval class and are only offered in agent mode — tools/registry.py).

No MCP servers exist in this offline environment; the protocol client is
tested against an in-repo echo server (tests/test_mcp.py).
"""

from __future__ import annotations

import json
import os
import subprocess
import threading
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

MCP_PROTOCOL_VERSION = "2024-11-05"


@dataclass
class MCPTool:
    name: str
    description: str
    input_schema: Dict[str, Any]
    server: str


class MCPServerConnection:
    """One stdio JSON-RPC connection to an MCP server process."""

    def __init__(self, name: str, command: List[str], env: Optional[Dict[str, str]] = None,
                 timeout: float = 20.0) -> None:
        self.name = name
        self.timeout = timeout
        self.proc = subprocess.Popen(
            command, stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            stderr=subprocess.DEVNULL, text=True, bufsize=1,
            env={**os.environ, **(env or {})})
        self._pending: Dict[str, Any] = {}
        self._lock = threading.Lock()
        self._events: Dict[str, threading.Event] = {}
        self._reader = threading.Thread(target=self._read_loop, daemon=True)
        self._reader.start()

    def _read_loop(self) -> None:
        try:
            for line in self.proc.stdout:
                line = line.strip()
                if not line:
                    continue
                try:
                    msg = json.loads(line)
                except ValueError:
                    continue
                rid = str(msg.get("id", ""))
                with self._lock:
                    self._pending[rid] = msg
                    ev = self._events.get(rid)
                if ev:
                    ev.set()
        except (OSError, ValueError):
            pass

    def request(self, method: str, params: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
        rid = str(uuid.uuid4())
        ev = threading.Event()
        with self._lock:
            self._events[rid] = ev
        payload = {"jsonrpc": "2.0", "id": rid, "method": method, "params": params or {}}
        self.proc.stdin.write(json.dumps(payload) + "\n")
        self.proc.stdin.flush()
        if not ev.wait(self.timeout):
            raise TimeoutError(f"MCP server {self.name}: no response to {method}")
        with self._lock:
            msg = self._pending.pop(rid)
            self._events.pop(rid, None)
        if "error" in msg:
            raise RuntimeError(f"MCP error from {self.name}: {msg['error']}")
        return msg.get("result", {})

    def initialize(self) -> Dict[str, Any]:
        result = self.request("initialize", {
            "protocolVersion": MCP_PROTOCOL_VERSION,
            "clientInfo": {"name": "senweaver_amd", "version": "0.1.0"},
            "capabilities": {},
        })
        # notification (no id, no reply expected)
        self.proc.stdin.write(json.dumps({"jsonrpc": "2.0",
                                          "method": "notifications/initialized"}) + "\n")
        self.proc.stdin.flush()
        return result

    def list_tools(self) -> List[MCPTool]:
        result = self.request("tools/list")
        return [MCPTool(t.get("name", ""), t.get("description", ""),
                        t.get("inputSchema", {}), self.name)
                for t in result.get("tools", [])]

    def call_tool(self, name: str, arguments: Dict[str, Any]) -> Any:
        result = self.request("tools/call", {"name": name, "arguments": arguments})
        content = result.get("content", [])
        texts = [c.get("text", "") for c in content if c.get("type") == "text"]
        return "\n".join(texts) if texts else result

    def close(self) -> None:
        try:
            self.proc.terminate()
            self.proc.wait(timeout=5)
        except (OSError, subprocess.TimeoutExpired):
            self.proc.kill()


class MCPService:
    """Loads mcp.json ({"mcpServers": {name: {command, args, env}}}) and
    aggregates every connected server's tools."""

    def __init__(self) -> None:
        self.connections: Dict[str, MCPServerConnection] = {}
        self.tools: List[MCPTool] = []

    def load_config(self, path: str) -> List[str]:
        try:
            data = json.load(open(path))
        except (OSError, ValueError):
            return []
        started = []
        for name, spec in (data.get("mcpServers") or {}).items():
            cmd = [spec.get("command", "")] + list(spec.get("args", []))
            try:
                self.connect(name, cmd, spec.get("env"))
                started.append(name)
            except Exception:
                continue
        return started

    def connect(self, name: str, command: List[str],
                env: Optional[Dict[str, str]] = None) -> List[MCPTool]:
        conn = MCPServerConnection(name, command, env)
        conn.initialize()
        tools = conn.list_tools()
        self.connections[name] = conn
        self.tools.extend(tools)
        return tools

    def tool_specs(self) -> List[dict]:
        """Shape consumed by tools.registry.available_tools(mcp_tools=...)."""
        return [{"name": t.name, "params": list(t.input_schema.get("properties", {}))}
                for t in self.tools]

    def call_mcp_tool(self, name: str, arguments: Dict[str, Any]) -> Any:
        for t in self.tools:
            if t.name == name:
                return self.connections[t.server].call_tool(name, arguments)
        raise KeyError(f"unknown MCP tool {name!r}")

    def close(self) -> None:
        for c in self.connections.values():
            c.close()
        self.connections.clear()
        self.tools.clear()
