"""Batched serving decode: aggregate tok/s at B concurrent sequences.

Lockstep greedy decode through the captured graph (DecodeGraph.step_batch):
one replay advances every sequence.  Weights stream once per STEP regardless
of B, so aggregate throughput scales until the B-row GEMV/attention stops
being weight-bound (~B=8-16 at 8B).
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd.engine.graph import DecodeGraph  # noqa: E402
from senweaver_amd.engine.kvcache import PAGE_SIZE, PagedKVCache  # noqa: E402
from senweaver_amd.models import LlamaModel, get_config  # noqa: E402
from senweaver_amd import ops  # noqa: E402


def main():
    quant = sys.argv[1] if len(sys.argv) > 1 else "bf16"
    cfg = get_config("llama-3-8b")
    model = LlamaModel(cfg, device="cuda:0", seed=3, quant=quant)
    print("quant:", quant)
    n = 48
    prompt_len = 128
    for B in (1, 2, 4, 8):
        pages_per_seq = (prompt_len + n) // PAGE_SIZE + 2
        cache = PagedKVCache(cfg, B * pages_per_seq + 2, torch.device("cuda:0"),
                             num_kv_heads=model.local_kv_heads)
        seqs = [cache.new_seq() for _ in range(B)]
        toks = torch.randint(0, cfg.vocab_size, (B, prompt_len),
                             generator=torch.Generator().manual_seed(7)).cuda()
        hidden = model.prefill(toks, cache=cache, seqs=seqs,
                               real_lens=[prompt_len] * B)
        last = hidden[:, prompt_len - 1]
        graph = DecodeGraph(model, cache, pages_per_seq + 1, batch=B)
        nxt = [int(x) for x in ops.argmax_rows(model.logits(last)).cpu()]
        for _ in range(4):  # warmup replays
            last = graph.step_batch(nxt, seqs)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            logits = model.logits(last)
            nxt = [int(x) for x in ops.argmax_rows(logits).cpu()]
            last = graph.step_batch(nxt, seqs)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"B={B}: {B * n} tokens in {dt:.2f}s = {B * n / dt:6.1f} tok/s "
              f"aggregate ({1000 * dt / n:.1f} ms/step)")


if __name__ == "__main__":
    main()
