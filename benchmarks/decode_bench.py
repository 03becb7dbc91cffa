"""Decode (textual-gradient generation) benchmark: tokens/s for greedy decode."""
import sys, time
import torch
sys.path.insert(0, ".")
from senweaver_amd.engine.scorer import LlamaBackend

def main():
    backend = LlamaBackend("llama-3-8b", max_seq=1024)
    # warmup + prefill
    t0 = time.perf_counter()
    out = backend.generate("warmup " * 50, max_new_tokens=4)
    torch.cuda.synchronize()
    print(f"init+warmup {time.perf_counter()-t0:.1f}s")
    n = 64
    t0 = time.perf_counter()
    out = backend.generate("- improve the rules for tool use and verification " * 10,
                           max_new_tokens=n)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ntok = len(out.split()) if out else n
    print(f"decode: {ntok} tokens in {dt:.2f}s = {ntok/dt:.1f} tok/s "
          f"({1000*dt/max(ntok,1):.1f} ms/tok)")

if __name__ == "__main__":
    main()
