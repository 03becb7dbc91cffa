"""Serving demo: the native C++ daemon + engine worker + streaming client.

Builds senweaver_daemon (C++ Unix-socket JSON-RPC control plane), starts
it with the tiny CPU model, streams a generation over the socket with
cumulative onText events, demonstrates the instant client-side abort,
then shuts down.  On an MI355X the same daemon serves Llama-3-8B
(SENWEAVER_MODEL=llama-3-8b) with the hipGraph decode path underneath.

Run: python examples/serving_demo.py          (no GPU needed, ~40 s)
"""
import os
import subprocess
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from senweaver_amd.server.client import DaemonClient, build_daemon


def main():
    sock = os.path.join(tempfile.mkdtemp(prefix="senweaver_srv_"), "daemon.sock")
    print("[1] building the native daemon (C++)...")
    bin_path = build_daemon()
    env = dict(os.environ, SENWEAVER_MODEL="tiny-debug", SENWEAVER_MAX_SEQ="128")
    proc = subprocess.Popen([bin_path, "--socket", sock], env=env)
    try:
        deadline = time.time() + 60
        while not os.path.exists(sock) and time.time() < deadline:
            time.sleep(0.1)
        c = DaemonClient(sock)
        assert c.ping(timeout=120), "engine worker did not come up"
        print("[2] daemon up, engine ready (pid", proc.pid, ")")

        texts = []
        done = threading.Event()
        c.send_llm_message(
            [{"role": "user", "content": "stream me something"}],
            on_text=lambda m: texts.append(m["fullText"]),
            on_final=lambda m: (texts.append(m["fullText"]), done.set()),
            on_error=lambda m: done.set(),
            max_new_tokens=8)
        assert done.wait(timeout=180)
        print(f"[3] streamed {len(texts)} cumulative events; final "
              f"{len(texts[-1])} chars: {texts[-1][:50]!r}")

        # instant client-side abort (the reference's abortRef contract)
        rid = c.send_llm_message([{"role": "user", "content": "long task"}],
                                 on_text=lambda m: None,
                                 on_final=lambda m: None,
                                 on_error=lambda m: None,
                                 max_new_tokens=64)
        t0 = time.perf_counter()
        c.abort(rid)
        print(f"[4] abort returned in {1000 * (time.perf_counter() - t0):.1f} ms "
              "(client-side instant; the worker stops decoding asynchronously)")
        st = c.stats()
        print(f"[5] engine usage: {st['totalRequests']} requests, "
              f"{st['totalInputTokens']} in / {st['totalOutputTokens']} out tokens")
        c.shutdown()
        c.close()
        proc.wait(timeout=30)
        print("[6] daemon shut down cleanly; demo ok")
    finally:
        if proc.poll() is None:
            proc.terminate()


if __name__ == "__main__":
    main()
