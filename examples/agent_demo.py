"""Agent-loop demo on CPU: the tiny local backbone drives a traced tool
turn end to end — user message -> XML tool call -> tool result fed back
-> assistant answer -> trace spans -> 9-dim reward.

Run: python examples/agent_demo.py        (no GPU needed, ~20 s)

The backbone is the random-init tiny-debug model, so its prose is noise;
what the demo shows is the MACHINERY: the streaming grammar, the tool
gate, trace collection and reward scoring all running over a real local
model exactly as they would over Llama-3-8B on an MI355X.
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from senweaver_amd.chat import ChatThreadService, GlobalSettings
from senweaver_amd.engine.scorer import LlamaBackend
from senweaver_amd.models import tiny_debug
from senweaver_amd.storage import MemoryStorage
from senweaver_amd.tools.service import ToolsService
from senweaver_amd.trace.collector import TraceCollector
from senweaver_amd.transport.service import LLMMessageService


class ToolScriptedBackend:
    """Wraps the real local backbone but scripts the FIRST reply to be a
    tool call, so the loop deterministically exercises the tool path (a
    random-init model virtually never emits valid XML on its own)."""

    def __init__(self):
        self.inner = LlamaBackend(tiny_debug(), device="cpu", max_seq=256)
        self.calls = 0

    def stream_generate(self, prompt, max_new_tokens, should_stop, on_chunk):
        self.calls += 1
        if self.calls == 1:
            text = ("Checking the workspace file "
                    "<read_file><uri>notes.txt</uri></read_file>")
            on_chunk(text)
            return text
        # later turns: the REAL model decodes (random weights -> noise,
        # but it is the genuine engine decode path)
        return self.inner.stream_generate(prompt, min(16, max_new_tokens),
                                          should_stop, on_chunk)


def main():
    ws = tempfile.mkdtemp(prefix="senweaver_demo_")
    with open(os.path.join(ws, "notes.txt"), "w") as f:
        f.write("the rollout failed on step 3; retry with smaller batch\n")

    backend = ToolScriptedBackend()
    tc = TraceCollector(storage=MemoryStorage())
    svc = ChatThreadService(LLMMessageService(backend), ToolsService(ws), tc,
                            settings=GlobalSettings(auto_approve={"built-in": True}),
                            sleep=lambda s: None)
    thread = svc.open_thread()
    svc.add_user_message_and_stream_response(thread.id, "what do the notes say?")

    msgs = svc.get_thread(thread.id).messages
    print(f"[1] turn ran: {len(msgs)} messages")
    for m in msgs:
        kind = m.role
        preview = str(getattr(m, "content", ""))[:60].replace("\n", " ")
        print(f"      {kind:18s} {preview}")

    traces = tc.get_all_traces()
    assert traces, "turn must be traced"
    t = traces[-1]
    print(f"[2] trace: {len(t.spans)} spans:", [s.type for s in t.spans][:6])
    fr = t.summary.final_reward
    if fr is None:
        tc.end_trace(t.trace_id)
        t = tc.get_all_traces()[-1]
        fr = t.summary.final_reward
    print(f"[3] finalReward = {fr}")
    print("demo ok")


if __name__ == "__main__":
    main()
