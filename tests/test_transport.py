"""LLMMessageService: streaming contract, instant abort, synchronous mode."""

import threading
import time

from senweaver_amd.transport import LLMMessageService
from senweaver_amd.transport.service import LLMChatMessage


class SlowBackend:
    """Streams one char at a time; optionally blocks until released."""

    def __init__(self, text, step_delay=0.0, gate=None):
        self.text = text
        self.step_delay = step_delay
        self.gate = gate  # threading.Event to wait on mid-stream

    def stream_generate(self, prompt, max_new_tokens, should_stop, on_chunk):
        acc = ""
        for i, ch in enumerate(self.text):
            if should_stop():
                return acc
            if self.gate is not None and i == 3:
                self.gate.wait(timeout=10)
            acc += ch
            on_chunk(acc)
            if self.step_delay:
                time.sleep(self.step_delay)
        return acc


def msgs(content="hi"):
    return [LLMChatMessage(role="user", content=content)]


def test_streaming_cumulative_contract_and_final():
    svc = LLMMessageService(SlowBackend("hello <think>why</think> world"))
    texts, finals = [], []
    done = threading.Event()
    svc.send_llm_message(
        msgs(), on_text=lambda **m: texts.append((m["full_text"], m["full_reasoning"])),
        on_final_message=lambda **m: (finals.append(m), done.set()),
        on_error=lambda **m: done.set())
    assert done.wait(5)
    assert finals and finals[0]["full_text"] == "hello  world"
    assert finals[0]["full_reasoning"] == "why"
    # cumulative: each full_text extends the previous
    for (a, _), (b, _) in zip(texts, texts[1:]):
        assert b.startswith(a)


def test_instant_abort_no_round_trip():
    """abort() resolves client-side immediately: the abort
    callback fires even while the backend is still blocked mid-stream."""
    gate = threading.Event()
    svc = LLMMessageService(SlowBackend("abcdefgh", gate=gate))
    aborted = threading.Event()
    got_final = threading.Event()
    rid = svc.send_llm_message(
        msgs(), on_text=lambda **m: None,
        on_final_message=lambda **m: got_final.set(),
        on_error=lambda **m: None,
        on_abort=lambda: aborted.set())
    time.sleep(0.05)     # let the stream reach the gate
    t0 = time.time()
    svc.abort(rid)       # returns instantly — no waiting on the backend
    assert time.time() - t0 < 0.1
    gate.set()           # backend resumes, sees should_stop, exits
    assert aborted.wait(5)
    assert not got_final.is_set()


def test_synchronous_mode_and_empty_error():
    svc = LLMMessageService(SlowBackend(""))
    errors = []
    svc.send_llm_message(msgs(), on_text=lambda **m: None,
                         on_final_message=lambda **m: None,
                         on_error=lambda **m: errors.append(m["message"]),
                         synchronous=True)
    assert errors and "empty" in errors[0].lower()


def test_wait_joins_thread():
    svc = LLMMessageService(SlowBackend("xyz", step_delay=0.01))
    done = threading.Event()
    rid = svc.send_llm_message(msgs(), on_text=lambda **m: None,
                               on_final_message=lambda **m: done.set(),
                               on_error=lambda **m: done.set())
    svc.wait(rid, timeout=5)
    assert done.is_set()


def test_model_options_passthrough():
    """model_options (temperature/topP) reach the backend's sampler."""
    import threading
    from senweaver_amd.transport.service import LLMMessageService

    seen = {}

    class SamplingBackend:
        def stream_generate(self, prompt, max_new_tokens, should_stop,
                            on_chunk, temperature=0.0, top_p=1.0,
                            sample_seed=None, stop=None):
            seen.update(temperature=temperature, top_p=top_p,
                        sample_seed=sample_seed)
            on_chunk("ok")
            return "ok"

    svc = LLMMessageService(SamplingBackend())
    done = threading.Event()
    svc.send_llm_message([], on_text=lambda **k: None,
                         on_final_message=lambda **k: done.set(),
                         on_error=lambda **k: done.set(),
                         model_options={"temperature": 0.7, "topP": 0.9,
                                        "sampleSeed": 42},
                         synchronous=True)
    assert done.is_set()
    assert seen == {"temperature": 0.7, "top_p": 0.9, "sample_seed": 42}


def test_render_messages_accepts_dicts():
    from senweaver_amd.transport.service import LLMChatMessage, LLMMessageService
    a = LLMMessageService.render_messages(
        [LLMChatMessage("user", "hi"), LLMChatMessage("assistant", "yo")])
    b = LLMMessageService.render_messages(
        [{"role": "user", "content": "hi"},
         {"role": "assistant", "content": "yo"}])
    assert a == b


def test_usage_tracking_per_request():
    from senweaver_amd.transport.service import LLMMessageService

    class EchoBackend:
        def stream_generate(self, prompt, max_new_tokens, should_stop,
                            on_chunk, **kw):
            on_chunk("four char out!")
            return "four char out!"

    svc = LLMMessageService(EchoBackend())
    svc.send_llm_message([{"role": "user", "content": "x" * 40}],
                         on_text=lambda **k: None,
                         on_final_message=lambda **k: None,
                         on_error=lambda **k: None, synchronous=True)
    st = svc.usage.stats()
    assert st["totalRequests"] == 1
    # 4 c/t estimate over the RENDERED prompt (role wrappers included)
    from senweaver_amd.transport.service import LLMChatMessage
    rendered = LLMMessageService.render_messages(
        [LLMChatMessage("user", "x" * 40)])
    assert st["totalInputTokens"] == len(rendered) // 4
    assert st["totalOutputTokens"] == len("four char out!") // 4
    assert st["byModel"]["local"]["requests"] == 1


def test_list_models_detailed_capabilities():
    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.transport.service import LLMMessageService
    svc = LLMMessageService(LlamaBackend("tiny-debug", device="cpu", max_seq=64))
    det = svc.list_models_detailed()
    assert det and det[0]["name"] == "tiny-debug"
    assert det[0]["contextWindow"] == 2048  # capabilities registry record
    assert det[0]["reservedOutputTokenSpace"] == 256


def test_generations_serialized_per_backend():
    """Concurrent sends must not interleave stream_generate on the shared
    backend (the paged cache/graph is single-stream state)."""
    import threading
    import time as _time
    from senweaver_amd.transport.service import LLMMessageService

    active = [0]
    max_active = [0]
    lock = threading.Lock()

    class SlowBackend:
        def stream_generate(self, prompt, max_new_tokens, should_stop,
                            on_chunk, **kw):
            with lock:
                active[0] += 1
                max_active[0] = max(max_active[0], active[0])
            _time.sleep(0.05)
            on_chunk("x")
            with lock:
                active[0] -= 1
            return "x"

    svc = LLMMessageService(SlowBackend())
    done = [threading.Event() for _ in range(5)]
    for i in range(5):
        svc.send_llm_message([{"role": "user", "content": str(i)}],
                             on_text=lambda **k: None,
                             on_final_message=lambda i=i, **k: done[i].set(),
                             on_error=lambda i=i, **k: done[i].set())
    assert all(d.wait(timeout=60) for d in done)
    assert max_active[0] == 1, f"generations overlapped: {max_active[0]}"


def test_penalties_reach_backend():
    import threading
    from senweaver_amd.transport.service import LLMMessageService

    seen = {}

    class B:
        def stream_generate(self, prompt, max_new_tokens, should_stop,
                            on_chunk, temperature=0.0, top_p=1.0,
                            sample_seed=None, stop=None,
                            presence_penalty=0.0, frequency_penalty=0.0):
            seen.update(pp=presence_penalty, fp=frequency_penalty)
            on_chunk("k")
            return "k"

    svc = LLMMessageService(B())
    svc.send_llm_message([{"role": "user", "content": "x"}],
                         on_text=lambda **k: None,
                         on_final_message=lambda **k: None,
                         on_error=lambda **k: None,
                         model_options={"temperature": 0.5,
                                        "presencePenalty": 0.3,
                                        "frequencyPenalty": 0.7},
                         synchronous=True)
    assert seen == {"pp": 0.3, "fp": 0.7}
