#!/usr/bin/env python3
"""End-to-end APO walkthrough on CPU — the full reference pipeline, locally.

Runs the complete loop a SenWeaver-IDE user gets from the RL subsystem:
  1. agent conversations are traced (TraceCollector spans),
  2. user feedback triggers the 9-dimension reward,
  3. the pattern detector + report builder find weaknesses,
  4. a textual gradient is generated and beam search optimizes the prompt
     (here with the tiny CPU model — on an MI355X the same code path runs
     Llama-3-8B on the HIP kernels; pass --model llama-3-8b on a GPU box),
  5. the winning rules are packed into the 2000-char budget and injected
     into the system message.

Everything persists through the same storage keys as the reference, so
state survives restart mid-search.

Usage: python examples/apo_demo.py [--model tiny-debug]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from senweaver_amd.apo import APOService, BeamSearchEngine
from senweaver_amd.apo.optimizer import LocalGradientEngine
from senweaver_amd.apo.rules import inject_rules
from senweaver_amd.engine.scorer import LlamaBackend
from senweaver_amd.storage import MemoryStorage
from senweaver_amd.trace import TraceCollector


def simulate_conversations(tc: TraceCollector) -> None:
    """A few traced agent turns with mixed outcomes + feedback."""
    for i in range(6):
        tid = tc.start_trace(f"th{i}", {"chatMode": "agent"})
        tc.record_user_message(f"th{i}", 0, f"fix the failing test #{i}")
        ok = i % 3 != 0
        for j in range(4):
            tc.record_tool_call(f"th{i}", j + 1, tool_name="edit_file",
                                tool_success=(ok or j % 2 == 0), duration=1200)
        tc.record_llm_call(f"th{i}", 5, input_tokens=4000, output_tokens=900,
                           duration=1500)
        tc.record_assistant_message(f"th{i}", 6, "done" if ok else "tried")
        tc.end_trace(tid)
        tc.record_user_feedback(f"th{i}", 6, "good" if ok else "bad")


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny-debug")
    args = ap.parse_args()

    backend = LlamaBackend(args.model, max_seq=512, micro_batch=2)
    engine = LocalGradientEngine(backend)
    storage = MemoryStorage()
    tc = TraceCollector(storage=storage)
    apo = APOService(tc, storage=storage, optimizer=engine)

    simulate_conversations(tc)
    traces = tc.get_all_traces()
    print(f"[1] {len(traces)} traces; rewards:",
          [round(t.summary.final_reward, 3) for t in traces])

    report = apo.analyze_prompt_effectiveness()
    print(f"[2] report: goodRate={report.good_rate:.2f}, "
          f"{len(report.patterns)} patterns, "
          f"{len(apo.get_pending_suggestions())} suggestions")
    for p in report.patterns[:3]:
        print("    -", p.description)

    grad = apo.request_textual_gradient()
    if grad is not None:
        print(f"[3] textual gradient ({grad.rollout_summary}):",
              grad.critique[:90].replace("\n", " "), "...")

    beam = BeamSearchEngine(backend)
    state = beam.run_search(apo, rounds=1)
    print(f"[4] beam round {state.current_round}/{state.total_rounds}: "
          f"best {state.history_best_score:.4f}, {len(state.beam)} in beam")

    rules = apo.get_optimized_rules()
    sysmsg = inject_rules("You are a careful coding agent.", rules)
    print(f"[5] {len(rules)} rules injected; system message {len(sysmsg)} chars")
    apo.flush(); tc.flush()
    print("\nstate persisted under:", sorted(storage._data.keys())[:4], "...")


if __name__ == "__main__":
    main()
