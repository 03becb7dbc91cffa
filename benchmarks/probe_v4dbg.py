"""Localize v4/v5 numerics failure: error pattern + determinism check."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from senweaver_amd import ops  # noqa: E402

dev = "cuda:0"
ext = ops.hip_ext()

for var in (23,):
    for (M, N, K) in [(512, 512, 512), (4096, 4096, 4096)]:
        g = torch.Generator().manual_seed(7)
        a = torch.randn(M, K, generator=g).bfloat16().to(dev)
        b = torch.randn(N, K, generator=g).bfloat16().to(dev)
        cb = (a.float() @ b.float().t())
        c1 = ext.gemm_bt_8ph_v(a, b, var).float()
        c2 = ext.gemm_bt_8ph_v(a, b, var).float()
        det = torch.equal(c1, c2)
        err = (c1 - cb).abs()
        tol = 0.5 + 0.02 * cb.abs()
        bad = err > tol
        nbad = int(bad.sum())
        print(f"v{var} {M}x{N}x{K}: bad {nbad}/{bad.numel()} deterministic={det} "
              f"maxerr={err.max().item():.2f}")
        if nbad:
            idx = bad.nonzero()
            r, c = idx[:, 0], idx[:, 1]
            # pattern within a 256-tile: fragment row/col bands + kk parity?
            print("   row%256//16 hist:", torch.bincount((r % 256) // 16, minlength=16).tolist())
            print("   col%256//16 hist:", torch.bincount((c % 256) // 16, minlength=16).tolist())
            print("   tile_m hist:", torch.bincount(r // 256).tolist()[:16])
            print("   tile_n hist:", torch.bincount(c // 256).tolist()[:16])
            samp = idx[:5]
            for (ri, ci) in samp.tolist():
                print(f"   [{ri},{ci}] got {c1[ri,ci]:.3f} want {cb[ri,ci]:.3f}")
