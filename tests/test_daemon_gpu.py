"""Daemon end-to-end on the GPU: C++ control plane streaming a REAL device
decode — the engine worker loads the HIP extension and
decodes on cuda:0; the client sees cumulative onText chunks and a final.
"""

import os
import subprocess
import threading
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

from senweaver_amd.server.client import DaemonClient, build_daemon


@pytest.fixture(scope="module")
def gpu_daemon():
    assert torch.cuda.is_available()
    bin_path = build_daemon()
    sock = f"/tmp/senweaver_gpu_{os.getpid()}.sock"
    env = dict(os.environ, SENWEAVER_MODEL="tiny-debug", SENWEAVER_MAX_SEQ="256")
    proc = subprocess.Popen([bin_path, "--socket", sock], env=env,
                            stderr=subprocess.PIPE)
    for _ in range(200):
        if os.path.exists(sock):
            break
        time.sleep(0.05)
    else:
        proc.kill()
        pytest.fail("daemon socket never appeared")
    yield sock
    proc.terminate()
    proc.wait(timeout=15)


def test_daemon_streams_gpu_decode(gpu_daemon):
    c = DaemonClient(gpu_daemon)
    assert c.ping(timeout=240)  # engine import + CUDA init
    texts = []
    done = threading.Event()
    c.send_llm_message(
        [{"role": "user", "content": "stream me a reply"}],
        on_text=lambda m: texts.append(m["fullText"]),
        on_final=lambda m: (texts.append(m["fullText"]), done.set()),
        on_error=lambda m: done.set(),
        max_new_tokens=12)
    assert done.wait(timeout=240), "no final message from GPU decode"
    assert texts and texts[-1]
    # cumulative contract: each onText extends the previous
    for a, b in zip(texts, texts[1:]):
        assert b.startswith(a[: len(a)])
