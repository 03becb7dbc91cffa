"""GEMV v2 (row-per-wave nt register pipeline) vs v3 (loader/consumer
LDS-DMA streaming engine) at the llama-8B decode shapes, M=1.
Effective TB/s = weight bytes / time (x and C are noise at these sizes).
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd import ops  # noqa: E402

dev = "cuda:0"
ext = ops.hip_ext()

shapes = [("qkv", 6144, 4096), ("o", 4096, 4096),
          ("gateup", 28672, 4096), ("down", 4096, 14336)]
for name, N, K in shapes:
    x = torch.randn(1, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    ref = (x.float() @ w.float().t())
    c2 = ext.gemm_bt(x, w)
    c3 = ext.gemv_bt_v3(x, w)
    r2 = ((c2.float() - ref).norm() / ref.norm()).item()
    r3 = ((c3.float() - ref).norm() / ref.norm()).item()
    ok3 = "ok" if r3 < 2e-2 else "WRONG"
    arms = {"v2": lambda: ext.gemm_bt(x, w), "v3": lambda: ext.gemv_bt_v3(x, w)}
    for fn in arms.values():
        for _ in range(10):
            fn()
    torch.cuda.synchronize()
    res = {}
    gb = N * K * 2
    for an, fn in arms.items():
        ts = []
        for _ in range(5):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(50):
                fn()
            torch.cuda.synchronize()
            ts.append((time.perf_counter() - t0) / 50)
        ts.sort()
        res[an] = gb / ts[len(ts) // 2] / 1e12
    print(f"{name:7s} N{N} K{K}: v2 {res['v2']:.2f} TB/s   v3 {res['v3']:.2f} TB/s"
          f"   (v3 rel={r3:.4f} {ok3}, v2 rel={r2:.4f})")
