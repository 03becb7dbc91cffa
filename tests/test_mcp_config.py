"""MCP client (against an in-repo echo server) + unified config system."""

import json
import sys
import textwrap

import pytest

from senweaver_amd.config import EngineConfig
from senweaver_amd.tools.mcp import MCPService
from senweaver_amd.tools.registry import available_tools

ECHO_SERVER = textwrap.dedent("""
    import json, sys
    for line in sys.stdin:
        msg = json.loads(line)
        m = msg.get("method")
        rid = msg.get("id")
        if rid is None:
            continue  # notification
        if m == "initialize":
            r = {"protocolVersion": "2024-11-05", "serverInfo": {"name": "echo"}}
        elif m == "tools/list":
            r = {"tools": [{"name": "echo_tool", "description": "echoes",
                            "inputSchema": {"type": "object",
                                            "properties": {"text": {"type": "string"}}}}]}
        elif m == "tools/call":
            t = msg["params"]["arguments"].get("text", "")
            r = {"content": [{"type": "text", "text": "echo: " + t}]}
        else:
            r = {}
        sys.stdout.write(json.dumps({"jsonrpc": "2.0", "id": rid, "result": r}) + "\\n")
        sys.stdout.flush()
""")


def test_mcp_handshake_and_call(tmp_path):
    server = tmp_path / "echo_server.py"
    server.write_text(ECHO_SERVER)
    svc = MCPService()
    tools = svc.connect("echo", [sys.executable, str(server)])
    try:
        assert tools and tools[0].name == "echo_tool"
        assert svc.tool_specs() == [{"name": "echo_tool", "params": ["text"]}]
        out = svc.call_mcp_tool("echo_tool", {"text": "hello"})
        assert out == "echo: hello"
        # MCP tools flow into the agent-mode tool list only
        agent = available_tools("agent", svc.tool_specs())
        assert any(t["name"] == "echo_tool" for t in agent)
        assert not any(t["name"] == "echo_tool"
                       for t in available_tools("gather", svc.tool_specs()))
    finally:
        svc.close()


def test_mcp_config_file(tmp_path):
    server = tmp_path / "echo_server.py"
    server.write_text(ECHO_SERVER)
    cfg = tmp_path / "mcp.json"
    cfg.write_text(json.dumps({
        "mcpServers": {
            "echo": {"command": sys.executable, "args": [str(server)]},
            "broken": {"command": "/nonexistent/bin"},
        }}))
    svc = MCPService()
    started = svc.load_config(str(cfg))
    try:
        assert started == ["echo"]  # broken server skipped gracefully
    finally:
        svc.close()


def test_engine_config_defaults_and_overrides(tmp_path):
    cfg = EngineConfig.load(env={})
    # reference defaults verbatim
    assert cfg.apo["beamWidth"] == 4 and cfg.apo["beamRounds"] == 3
    assert cfg.trace["maxTraces"] == 1000 and cfg.trace["maxSpansPerTrace"] == 200
    assert cfg.context["apoRulesMaxChars"] == 2000
    assert cfg.context["charsPerToken"] == 3.5
    # file + env overrides
    f = tmp_path / "engine.json"
    f.write_text(json.dumps({"apo": {"beamWidth": 8}, "engine": {"model": "llama-3-70b"}}))
    cfg2 = EngineConfig.load(str(f), env={"SENWEAVER_APO_BRANCHFACTOR": "2",
                                          "SENWEAVER_TRACE_MAXTRACES": "500",
                                          "SENWEAVER_ENGINE_QUANT": "fp8",
                                          "SENWEAVER_APO_AUTOANALYZEENABLED": "false"})
    assert cfg2.apo["beamWidth"] == 8
    assert cfg2.apo["branchFactor"] == 2
    assert cfg2.apo["autoAnalyzeEnabled"] is False
    assert cfg2.trace["maxTraces"] == 500
    assert cfg2.engine["model"] == "llama-3-70b"
    assert cfg2.engine["quant"] == "fp8"


def test_engine_config_bad_file(tmp_path):
    f = tmp_path / "bad.json"
    f.write_text("{not json")
    with pytest.raises(ValueError):
        EngineConfig.load(str(f), env={})


def test_trace_collector_thread_safety():
    """the reference is single-threaded by construction; our
    collector must survive concurrent fire-and-forget recording."""
    import threading
    from senweaver_amd.trace import TraceCollector

    tc = TraceCollector()

    def worker(tid):
        th = f"th{tid}"
        trace_id = tc.start_trace(th, {"chatMode": "agent"})
        for i in range(50):
            tc.record_user_message(th, i, f"msg {i}")
            tc.record_tool_call(th, i, tool_name="read_file", tool_success=True, duration=5)
        tc.end_trace(trace_id)
        tc.record_user_feedback(th, 1, "good")

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    stats = tc.get_stats()
    assert stats["totalTraces"] == 8
    assert stats["goodFeedbacks"] == 8
    assert stats["totalToolCalls"] == 8 * 50
    for trace in tc.get_all_traces():
        assert trace.summary.final_reward is not None
