"""Within-probe interleaved A/B of the 8-phase GEMM structural variants.

Variants (gemm_pipe.hip): 0 = 2 barriers/phase (template as written),
1 = 1 barrier/phase, 2 = V0 without setprio, 3 = V1 + static young-half
prio.  Baselines: old one-barrier-per-K-tile 256-tile kernel + hipBLASLt.
Interleaved rounds in one process (methodology rule 24); random data
(rule 25).  Pass "pmc" to run ONLY variant 0 a few times for a rocprofv3
--pmc pass.
"""
import os
import sys
import time

import torch


def timeit(fn, warm=3, it=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd import ops  # noqa: E402

dev = "cuda:0"
ext = ops.hip_ext()

pmc_mode = len(sys.argv) > 1 and sys.argv[1] == "pmc"

shapes = [(4096, 4096, 4096)] if pmc_mode else [(4096, 4096, 4096),
                                                (8192, 8192, 8192),
                                                (2048, 28672, 4096)]

for (M, N, K) in shapes:
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    fl = 2 * M * N * K / 1e12

    if pmc_mode:
        # a handful of dispatches of the leading variants + blas for counters
        for _ in range(3):
            ext.gemm_bt_8ph_v(a, b, 21)
        for _ in range(3):
            a @ b.t()
        torch.cuda.synchronize()
        print("pmc dispatches done")
        break

    arms = {

        "v14-16n": lambda: ext.gemm_bt_8ph_v(a, b, 14),
        "v18-asm": lambda: ext.gemm_bt_8ph_v(a, b, 18),
        "v21-pri": lambda: ext.gemm_bt_8ph_v(a, b, 21),
        "v24-gls": lambda: ext.gemm_bt_8ph_v(a, b, 24),
        "blas   ": lambda: a @ b.t(),
    }
    # numerics check each variant once vs blas (a wrong arm is dropped,
    # not fatal - this is a probe)
    cb = (a @ b.t()).float()
    for name in list(arms):
        c = arms[name]().float()
        err = (c - cb).abs().max().item()
        if err > max(1.5, 0.01 * cb.abs().max().item()):
            print(f"{name}: WRONG (maxabs {err:.3f}) - dropped")
            del arms[name]
    # warmup
    for fn in arms.values():
        for _ in range(3):
            fn()
    torch.cuda.synchronize()
    # interleaved rounds
    times = {k: [] for k in arms}
    for rnd in range(8):
        for name, fn in arms.items():
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(5):
                fn()
            torch.cuda.synchronize()
            times[name].append((time.perf_counter() - t0) / 5)
    line = f"{M}x{N}x{K}:"
    for name, ts in times.items():
        ts.sort()
        med = ts[len(ts) // 2]
        line += f"  {name} {fl / med:7.1f}/{fl / ts[0]:7.1f}"
    print(line + "   (TF/s med/max)")

# MX fp8 pipeline A/B (appended): pipe kernel vs old 256 vs _scaled_mm
if not pmc_mode:
    for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192), (2048, 28672, 4096)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        amq, ams = ops.quant_mxfp8(a)
        bmq, bms = ops.quant_mxfp8(b)
        fl = 2 * M * N * K / 1e12
        c_new = ops.gemm_bt_mxfp8(amq, ams, bmq, bms).float()
        c_ref = (a.float() @ b.float().t())
        err = (c_new - c_ref).abs().max().item()
        rel = ((c_new - c_ref).norm() / c_ref.norm()).item()
        t_new = timeit(lambda: ops.gemm_bt_mxfp8(amq, ams, bmq, bms))
        print(f"MXpipe {M}x{N}x{K}: {fl / t_new:7.1f} TF/s  rel={rel:.4f} maxerr={err:.2f}")
