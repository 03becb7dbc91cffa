"""Interleaved A/B of the attention v4 lever matrix (attention_v4.hip).

Variants: 0 v3-anchor, 1 +defer-max, 2 async-STAGE single-buf,
3 2+defer, 4 reg-stage double-buf pre-PV, 5 4+defer, 6 glds-after-QK,
7 6+defer.  Numerics vs the fp32 torch reference first (any wrong variant
is dropped from timing), then interleaved rounds on random data
(methodology rules 24/25) at the bench shapes.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd import ops  # noqa: E402
from senweaver_amd.ops import reference as ref  # noqa: E402

dev = "cuda:0"
ext = ops.hip_ext()
VARS = [0, 1, 3, 5, 11, 12, 13, 14, 15]
if len(sys.argv) > 1 and sys.argv[1] == "pmc":
    # counter run: a few dispatches of anchor + winner only, bench shape
    dev = "cuda:0"
    D = 128
    q = torch.randn(4, 32, 2048, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(4, 8, 2048, D, dtype=torch.bfloat16, device=dev)
    vt = torch.randn(4, 8, D, 2048, dtype=torch.bfloat16, device=dev)
    for var in (0, 15):
        for _ in range(4):
            ext.attn_fwd_v4(q, k, vt, D ** -0.5, var)
    torch.cuda.synchronize()
    print("pmc dispatches done")
    sys.exit(0)

# ---- numerics gate ----
B, H, Hk, S, D = 2, 8, 2, 1024, 128
g = torch.Generator(device="cpu").manual_seed(7)
q = torch.randn(B, H, S, D, generator=g, dtype=torch.bfloat16).to(dev)
k = torch.randn(B, Hk, S, D, generator=g, dtype=torch.bfloat16).to(dev)
v = torch.randn(B, Hk, S, D, generator=g, dtype=torch.bfloat16).to(dev)
vt = v.transpose(-1, -2).contiguous()
scale = D ** -0.5
want = ref.attn_fwd_ref(q, k, v, scale, causal=True)  # [B,H,S,D] fp32 path
ok_vars = []
for var in VARS:
    ot = ext.attn_fwd_v4(q, k, vt, scale, var)
    got = ot.transpose(-1, -2)  # [B,H,S,D]
    rel = ((got.float() - want.float()).norm() / want.float().norm()).item()
    mx = (got.float() - want.float()).abs().max().item()
    status = "ok" if rel < 2e-2 and mx < 0.1 else "WRONG"
    print(f"v{var}: rel={rel:.5f} maxabs={mx:.4f} {status}")
    if status == "ok":
        ok_vars.append(var)

# ---- perf: interleaved rounds ----
shapes = [(4, 32, 8, 2048), (1, 32, 8, 8192)]
for (B, H, Hk, S) in shapes:
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hk, S, D, dtype=torch.bfloat16, device=dev)
    vt = v.transpose(-1, -2).contiguous()
    flops = 2 * 2 * B * H * S * S * D / 2  # causal-half accounting
    arms = {var: [] for var in ok_vars}
    for var in ok_vars:  # warmup
        for _ in range(3):
            ext.attn_fwd_v4(q, k, vt, scale, var)
    torch.cuda.synchronize()
    for rnd in range(5):
        for var in ok_vars:
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                ext.attn_fwd_v4(q, k, vt, scale, var)
            torch.cuda.synchronize()
            arms[var].append((time.perf_counter() - t0) / 10)
    print(f"--- B{B} H{H} S{S} (causal TF/s, med of 5 rounds) ---")
    for var in ok_vars:
        ts = sorted(arms[var])
        med = ts[len(ts) // 2]
        print(f"v{var}: {flops / med / 1e12:7.1f} TF  (best {flops / ts[0] / 1e12:7.1f})")
