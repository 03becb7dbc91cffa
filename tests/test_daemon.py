"""Native daemon end-to-end: C++ JSON-RPC control plane + Python engine worker."""

import os
import subprocess
import threading
import time
import uuid

import pytest

from senweaver_amd.server.client import DaemonClient, build_daemon
from senweaver_amd.utils.cache import LRUCache, MultiLayerCache
from senweaver_amd.utils.observability import (
    MetricsService,
    PerformanceMonitor,
    TokenUsageTracker,
)


@pytest.fixture(scope="module")
def daemon():
    bin_path = build_daemon()
    sock = f"/tmp/senweaver_test_{os.getpid()}.sock"
    env = dict(os.environ, SENWEAVER_MODEL="tiny-debug", SENWEAVER_MAX_SEQ="128")
    proc = subprocess.Popen([bin_path, "--socket", sock], env=env,
                            stderr=subprocess.PIPE)
    # wait for the socket
    for _ in range(200):
        if os.path.exists(sock):
            break
        time.sleep(0.05)
    else:
        proc.kill()
        pytest.fail("daemon socket never appeared")
    yield sock, proc.pid
    proc.terminate()
    proc.wait(timeout=10)


def test_daemon_ping(daemon):
    c = DaemonClient(daemon[0])
    assert c.ping(timeout=120)  # first ping waits for engine import/ready
    c.close()


def test_daemon_stream_and_final(daemon):
    c = DaemonClient(daemon[0])
    assert c.ping(timeout=120)
    texts = []
    done = threading.Event()

    rid = c.send_llm_message(
        [{"role": "user", "content": "hello engine"}],
        on_text=lambda m: texts.append(m["fullText"]),
        on_final=lambda m: (texts.append(m["fullText"]), done.set()),
        on_error=lambda m: done.set(),
        max_new_tokens=6)
    assert done.wait(timeout=180), "no final message"
    # cumulative streaming: each onText extends the previous
    for a, b in zip(texts, texts[1:]):
        assert b.startswith(a[: len(a)])
    assert texts[-1]  # non-empty final
    c.close()


def test_daemon_abort(daemon):
    c = DaemonClient(daemon[0])
    assert c.ping(timeout=120)
    rid = c.send_llm_message([{"role": "user", "content": "long task"}],
                             max_new_tokens=64)
    c.abort(rid)
    time.sleep(1.0)
    aborts = [e for e in c.events if e.get("event") == "onAbort" and e.get("requestId") == rid]
    assert aborts, "abort ack not received"
    # no final message may arrive for the aborted id after the abort ack
    finals = [e for e in c.events if e.get("event") == "onFinalMessage" and e.get("requestId") == rid]
    assert not finals
    c.close()


# ---- aux utils (cache / observability) ----

def test_lru_cache():
    c = LRUCache(capacity=2)
    c.put("a", 1)
    c.put("b", 2)
    assert c.get("a") == 1
    c.put("c", 3)  # evicts b (a was touched)
    assert c.get("b") is None
    assert c.get("a") == 1 and c.get("c") == 3
    assert c.hit_rate is not None


def test_multilayer_cache():
    m = MultiLayerCache(hot_capacity=1, warm_capacity=4)
    m.put("x", 10)
    m.put("y", 20)  # x falls out of hot, stays warm
    assert m.get("x") == 10  # warm hit promotes back


def test_metrics_and_usage():
    ms = MetricsService()
    ms.capture("llm_send", {"model": "local"})
    ms.capture("llm_send", {"model": "local"})
    assert ms.debug_info()["byEvent"]["llm_send"] == 2
    t = TokenUsageTracker()
    t.record("r1", "llama-3-8b", 100, 50)
    t.record("r2", "llama-3-8b", 10, 5)
    s = t.stats()
    assert s["totalInputTokens"] == 110
    assert s["byModel"]["llama-3-8b"]["requests"] == 2


def test_performance_monitor_slo():
    pm = PerformanceMonitor(enabled=True)
    pm.record("messageTrimming", 500)  # over the 200ms SLO
    pm.record("messageTrimming", 10)
    assert pm.violations and pm.violations[0]["metric"] == "messageTrimming"
    assert pm.summary()["messageTrimming"]["count"] == 2
    with pm.timer("systemMessageGeneration"):
        pass
    assert pm.summary()["systemMessageGeneration"]["count"] == 1


def test_daemon_worker_crash_restart(daemon):
    """Kill the engine worker mid-session: the daemon detects the exit,
    respawns it (daemon.cpp waitpid/restart path), and subsequent requests
    are served by the fresh worker."""
    import signal

    sock, daemon_pid = daemon
    c = DaemonClient(sock)
    assert c.ping(timeout=120)
    # the worker is THE child of OUR daemon process (exact pid, no patterns)
    out = subprocess.run(["pgrep", "-P", str(daemon_pid)], capture_output=True,
                         text=True)
    pids = [int(x) for x in out.stdout.split()]
    assert pids, "engine worker child process not found"
    os.kill(pids[0], signal.SIGKILL)

    done = threading.Event()
    texts = []
    deadline = time.time() + 180
    while time.time() < deadline:
        done.clear()
        texts.clear()
        c.send_llm_message([{"role": "user", "content": "after restart"}],
                           on_final=lambda m: (texts.append(m["fullText"]), done.set()),
                           on_error=lambda m: done.set(),
                           max_new_tokens=4)
        if done.wait(timeout=60) and texts:
            break
        time.sleep(1.0)
    assert texts, "no successful response after worker restart"
    c.close()


def test_daemon_worker_crash_errors_inflight(daemon):
    """An in-flight request whose worker dies must receive onError (not hang):
    daemon.cpp errors out every outstanding requestId before respawning, the
    same contract as the reference channel erroring the request back."""
    import signal

    sock, daemon_pid = daemon
    c = DaemonClient(sock)
    assert c.ping(timeout=120)
    out = subprocess.run(["pgrep", "-P", str(daemon_pid)], capture_output=True,
                         text=True)
    pids = [int(x) for x in out.stdout.split()]
    assert pids, "engine worker child process not found"

    done = threading.Event()
    errors = []
    rid = c.send_llm_message([{"role": "user", "content": "doomed request"}],
                             on_final=lambda m: done.set(),
                             on_error=lambda m: (errors.append(m), done.set()),
                             max_new_tokens=512)
    time.sleep(0.3)  # let the request reach the worker
    os.kill(pids[0], signal.SIGKILL)
    assert done.wait(timeout=60), "in-flight request hung after worker crash"
    assert errors and "crash" in errors[0].get("message", "")
    c.close()


def test_cli_singleton_start_status_stop(tmp_path, monkeypatch):
    """Launcher CLI (server/cli.py = the Rust code-cli's engine subset):
    singleton lock, start/status/stop, stale-lock reclaim."""
    from senweaver_amd.server import cli

    monkeypatch.setattr(cli, "STATE_DIR", str(tmp_path))
    monkeypatch.setattr(cli, "SOCKET_PATH", str(tmp_path / "d.sock"))
    monkeypatch.setattr(cli, "LOCK_PATH", str(tmp_path / "d.lock"))
    monkeypatch.setattr(cli, "LOG_PATH", str(tmp_path / "d.log"))
    monkeypatch.setenv("SENWEAVER_MODEL", "tiny-debug")
    monkeypatch.setenv("SENWEAVER_MAX_SEQ", "128")

    assert cli.read_lock() is None
    assert cli.start() == 0
    pid = cli.read_lock()
    assert pid is not None
    # second start is a no-op against the live singleton
    assert cli.start() == 0
    assert cli.read_lock() == pid
    assert cli.status() == 0
    assert cli.stop() == 0
    assert cli.read_lock() is None
    # stale-lock reclaim: dead pid in the lock file
    (tmp_path / "d.lock").write_text("999999")
    assert cli.read_lock() is None
    assert not (tmp_path / "d.lock").exists()


def test_daemon_slow_consumer_does_not_block_others(daemon):
    """A client that sends a generation request and then never reads its
    socket must not stall the daemon: events for it queue in its bounded
    per-client outbuf (daemon.cpp kMaxClientOutbuf) while OTHER clients
    keep getting full service."""
    import json
    import socket as socketlib

    sock_path = daemon[0]
    slow = socketlib.socket(socketlib.AF_UNIX, socketlib.SOCK_STREAM)
    slow.connect(sock_path)
    slow.sendall((json.dumps({
        "method": "sendLLMMessage", "requestId": "slow-1",
        "messages": [{"role": "user", "content": "slow consumer"}],
        "maxNewTokens": 24}) + "\n").encode())
    # deliberately never read from `slow`
    try:
        healthy = DaemonClient(sock_path)
        assert healthy.ping(timeout=120)
        texts = []
        done = threading.Event()
        healthy.send_llm_message(
            [{"role": "user", "content": "healthy client"}],
            on_text=lambda m: texts.append(m["fullText"]),
            on_final=lambda m: (texts.append(m["fullText"]), done.set()),
            on_error=lambda m: done.set(),
            max_new_tokens=4)
        assert done.wait(timeout=180), "healthy client starved by slow one"
        assert texts and texts[-1]
        healthy.close()
    finally:
        slow.close()


def test_daemon_model_options_sampling(daemon):
    """modelOptions (temperature, seed) flow client -> daemon -> worker ->
    backend sampler: two identical seeded requests stream identical text;
    a different seed explores differently (tiny random-init model)."""
    c = DaemonClient(daemon[0])
    assert c.ping(timeout=120)

    def run(seed):
        texts = []
        done = threading.Event()
        c.send_llm_message([{"role": "user", "content": "sample"}],
                           on_final=lambda m: (texts.append(m["fullText"]),
                                               done.set()),
                           on_error=lambda m: done.set(),
                           max_new_tokens=8,
                           model_options={"temperature": 1.0,
                                          "sampleSeed": seed})
        assert done.wait(timeout=180)
        return texts[0] if texts else ""

    a1, a2 = run(7), run(7)
    assert a1 == a2 and a1
    outs = {run(sd) for sd in range(5)}
    assert len(outs) > 1
    c.close()


def test_cli_one_shot_generate(tmp_path):
    """`senweaver-daemon generate` front door: auto-start, stream to
    stdout, daemon reusable after, clean stop."""
    import subprocess
    import sys
    env = dict(os.environ,
               SENWEAVER_STATE_DIR=str(tmp_path),
               SENWEAVER_SOCKET=str(tmp_path / "d.sock"),
               SENWEAVER_MODEL="tiny-debug", SENWEAVER_MAX_SEQ="128")
    r = subprocess.run([sys.executable, "-m", "senweaver_amd.server.cli",
                        "generate", "--max-new", "6", "one shot"],
                       capture_output=True, text=True, env=env, timeout=420)
    assert r.returncode == 0, r.stderr[-500:]
    assert r.stdout.strip()  # streamed something
    # seeded sampling through the CLI reuses the running daemon
    r_s = subprocess.run([sys.executable, "-m", "senweaver_amd.server.cli",
                          "generate", "--max-new", "6", "--temperature",
                          "1.0", "--seed", "3", "sampled shot"],
                         capture_output=True, text=True, env=env, timeout=420)
    assert r_s.returncode == 0, r_s.stderr[-500:]
    assert r_s.stdout.strip()
    r2 = subprocess.run([sys.executable, "-m", "senweaver_amd.server.cli",
                         "stop"], capture_output=True, text=True, env=env,
                        timeout=120)
    assert r2.returncode == 0


def test_daemon_crash_loop_breaker(tmp_path):
    """A worker that can never start (bogus model) must NOT respawn
    forever: after 3 fast crashes the daemon exits, the socket is gone,
    and the launcher reports it down."""
    bin_path = build_daemon()
    sock = str(tmp_path / "d.sock")
    env = dict(os.environ, SENWEAVER_MODEL="no-such-model")
    proc = subprocess.Popen([bin_path, "--socket", sock], env=env,
                            stderr=subprocess.PIPE, text=True)
    try:
        proc.wait(timeout=240)  # 3 failed spawns, then give up
    except subprocess.TimeoutExpired:
        proc.kill()
        pytest.fail("daemon kept respawning a dead worker")
    err = proc.stderr.read()
    assert "giving up" in err, err[-500:]
    assert not os.path.exists(sock)


def test_daemon_stats_rpc(daemon):
    c = DaemonClient(daemon[0])
    assert c.ping(timeout=120)
    done = threading.Event()
    c.send_llm_message([{"role": "user", "content": "count me"}],
                       on_final=lambda m: done.set(),
                       on_error=lambda m: done.set(), max_new_tokens=4)
    assert done.wait(timeout=180)
    st = c.stats()
    assert st and st["totalRequests"] >= 1
    assert st["totalInputTokens"] > 0
    c.close()


def test_agent_loop_over_daemon(daemon, tmp_path):
    """The full agent tool loop with its LLM calls served by the NATIVE
    daemon (the reference's chat-service <-> native-channel topology):
    DaemonBackend proxies stream_generate over the socket with rawPrompt
    (no double role-rendering)."""
    from senweaver_amd.chat import ChatThreadService, GlobalSettings
    from senweaver_amd.server.client import DaemonBackend
    from senweaver_amd.storage import MemoryStorage
    from senweaver_amd.tools.service import ToolsService
    from senweaver_amd.trace.collector import TraceCollector
    from senweaver_amd.transport.service import LLMMessageService

    backend = DaemonBackend(daemon[0])
    try:
        tc = TraceCollector(storage=MemoryStorage())
        svc = ChatThreadService(LLMMessageService(backend),
                                ToolsService(str(tmp_path)), tc,
                                settings=GlobalSettings(auto_approve={}),
                                sleep=lambda s: None)
        thread = svc.open_thread()
        svc.add_user_message_and_stream_response(thread.id, "say anything")
        msgs = svc.get_thread(thread.id).messages
        assert any(m.role == "assistant" for m in msgs)
        assert tc.get_all_traces()  # turn traced end to end
    finally:
        backend.close()


def test_daemon_many_concurrent_clients(daemon):
    """8 clients streaming simultaneously: every request completes, every
    stream stays cumulative, no cross-request text bleed."""
    results = {}
    lock = threading.Lock()

    def one(i):
        c = DaemonClient(daemon[0])
        texts = []
        done = threading.Event()
        c.send_llm_message([{"role": "user", "content": f"client {i}"}],
                           on_text=lambda m: texts.append(m["fullText"]),
                           on_final=lambda m: (texts.append(m["fullText"]),
                                               done.set()),
                           on_error=lambda m: done.set(),
                           max_new_tokens=5)
        ok = done.wait(timeout=300)
        for a, b in zip(texts, texts[1:]):
            assert b.startswith(a)
        with lock:
            results[i] = ok and bool(texts)
        c.close()

    threads = [threading.Thread(target=one, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=360)
    assert len(results) == 8 and all(results.values()), results


def test_http_over_daemon(daemon):
    """The composed topology: OpenAI-compatible HTTP -> DaemonBackend ->
    C++ daemon -> engine worker."""
    from fastapi.testclient import TestClient
    from senweaver_amd.server.client import DaemonBackend
    from senweaver_amd.server.http_api import create_app
    from senweaver_amd.transport.service import LLMMessageService

    backend = DaemonBackend(daemon[0])
    try:
        app = create_app(LLMMessageService(backend))
        with TestClient(app) as c:
            m = c.get("/v1/models").json()["data"]
            assert m and m[0]["id"] == "tiny-debug"  # names proxied over UDS
            r = c.post("/v1/completions",
                       json={"prompt": "through every layer", "max_tokens": 5})
            assert r.status_code == 200
            assert isinstance(r.json()["choices"][0]["text"], str)
    finally:
        backend.close()
