"""Direct stdio-protocol test of server/engine_worker.py (the model-side
half of the native daemon): spawn it as a subprocess with the tiny CPU
model and drive the newline-JSON request/event protocol end to end."""

import json
import os
import subprocess
import sys
import threading
import time

import pytest

WORKER = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "senweaver_amd", "server", "engine_worker.py")


@pytest.fixture(scope="module")
def worker():
    env = dict(os.environ, SENWEAVER_MODEL="tiny-debug", SENWEAVER_MAX_SEQ="128")
    p = subprocess.Popen([sys.executable, WORKER], stdin=subprocess.PIPE,
                         stdout=subprocess.PIPE, text=True, env=env,
                         bufsize=1)
    events = []
    cond = threading.Condition()

    def reader():
        for line in p.stdout:
            with cond:
                events.append(json.loads(line))
                cond.notify_all()

    t = threading.Thread(target=reader, daemon=True)
    t.start()

    def wait_for(pred, timeout=120):
        deadline = time.time() + timeout
        with cond:
            while time.time() < deadline:
                for e in events:
                    if pred(e):
                        return e
                cond.wait(timeout=1.0)
        raise AssertionError(f"no matching event; got {events[-5:]}")

    def send(obj):
        p.stdin.write(json.dumps(obj) + "\n")
        p.stdin.flush()

    yield send, wait_for, events, p
    send({"method": "shutdown"})
    try:
        p.wait(timeout=30)
    except subprocess.TimeoutExpired:
        p.kill()


def test_worker_ready_and_ping(worker):
    send, wait_for, _, _p = worker
    wait_for(lambda e: e.get("event") == "ready" and e.get("model") == "tiny-debug")
    send({"method": "ping"})
    wait_for(lambda e: e.get("event") == "pong")


def test_worker_list_models(worker):
    send, wait_for, _, _p = worker
    send({"method": "list"})
    e = wait_for(lambda e: e.get("event") == "listResult")
    assert any("tiny" in m or "llama" in m for m in e["models"])


def test_worker_generation_streams_cumulative(worker):
    send, wait_for, events, _p = worker
    send({"method": "sendLLMMessage", "requestId": "rq1", "maxNewTokens": 6,
          "messages": [{"role": "user", "content": "hello worker"}]})
    final = wait_for(lambda e: e.get("event") == "onFinalMessage"
                     and e.get("requestId") == "rq1", timeout=180)
    texts = [e["fullText"] for e in events
             if e.get("event") == "onText" and e.get("requestId") == "rq1"]
    # cumulative contract: each onText extends the previous
    for a, b in zip(texts, texts[1:]):
        assert b.startswith(a)
    if texts:
        assert final["fullText"].startswith(texts[-1])


def test_worker_bad_json_is_nonfatal(worker):
    send, wait_for, events, p = worker
    p.stdin.write("{not json}\n")
    p.stdin.flush()
    wait_for(lambda e: e.get("event") == "onError"
             and e.get("message") == "bad json")
    # the worker keeps serving after the bad line
    pongs0 = sum(1 for e in events if e.get("event") == "pong")
    send({"method": "ping"})
    wait_for(lambda e: sum(1 for x in events if x.get("event") == "pong") > pongs0)
