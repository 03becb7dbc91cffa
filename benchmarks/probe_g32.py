import sys, time
import torch
sys.path.insert(0, ".")
from senweaver_amd import ops

dev = "cuda:0"
ext = ops.hip_ext()

def timeit(fn, warm=5, it=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it

for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192), (2048, 28672, 4096),
                  (2048, 128256, 4096)]:
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    Mp = (M + 255) // 256 * 256
    ap = torch.nn.functional.pad(a, (0, 0, 0, Mp - M)) if Mp != M else a
    Np = (N + 255) // 256 * 256
    bp = torch.nn.functional.pad(b, (0, 0, 0, Np - N)) if Np != N else b
    c32 = ext.gemm_bt_256x32(ap, bp)[:M, :N]
    csg = ext.gemm_bt_256sg(ap, bp)[:M, :N]
    c16 = ops.gemm_bt_tiled(a, b)
    rel = ((c32.float() - c16.float()).norm() / c16.float().norm()).item()
    relsg = ((csg.float() - c16.float()).norm() / c16.float().norm()).item()
    t32 = timeit(lambda: ext.gemm_bt_256x32(ap, bp))
    tsg = timeit(lambda: ext.gemm_bt_256sg(ap, bp))
    t16 = timeit(lambda: ops.gemm_bt_tiled(a, b))
    tbl = timeit(lambda: a @ b.t())
    fl = 2 * M * N * K / 1e12
    print(f"{M}x{N}x{K}: 16x16 {fl/t16:7.1f}  32x32 {fl/t32:7.1f}  sg {fl/tsg:7.1f}  blas {fl/tbl:7.1f} TF/s  rel={rel:.5f}/{relsg:.5f}")
