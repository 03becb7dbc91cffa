"""Decode (textual-gradient generation) benchmark: tokens/s for greedy decode."""
import sys, time
import torch
import os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd.engine.scorer import LlamaBackend

def main():
    quant = sys.argv[1] if len(sys.argv) > 1 else "bf16"
    backend = LlamaBackend("llama-3-8b", max_seq=1024, quant=quant)
    print("quant:", quant)
    # warmup + prefill
    t0 = time.perf_counter()
    out = backend.generate("warmup " * 50, max_new_tokens=4)
    torch.cuda.synchronize()
    print(f"init+warmup {time.perf_counter()-t0:.1f}s")
    n = 64
    count = [0]
    t0 = time.perf_counter()
    out = backend.stream_generate(
        "- improve the rules for tool use and verification " * 10,
        max_new_tokens=n, should_stop=lambda: False,
        on_chunk=lambda _t: count.__setitem__(0, count[0] + 1))
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ntok = max(count[0], 1)
    print(f"decode: {ntok} tokens in {dt:.2f}s = {ntok/dt:.1f} tok/s "
          f"({1000*dt/ntok:.1f} ms/tok)")
    print("sample:", out[:120].replace("\n", " "))

if __name__ == "__main__":
    main()
