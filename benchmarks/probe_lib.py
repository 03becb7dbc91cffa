import sys, time
import torch
sys.path.insert(0, ".")
from senweaver_amd import ops

dev = "cuda:0"
def timeit(fn, warm=5, it=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it

# ---- torch._scaled_mm fp8 (rowwise scales) ----
M, N, K = 8192, 8192, 8192
a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
b = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
aq, asc = ops.quant_fp8(a); bq, bsc = ops.quant_fp8(b)
af8 = aq.view(torch.float8_e4m3fn)
bf8 = bq.view(torch.float8_e4m3fn)
try:
    out = torch._scaled_mm(af8, bf8.t(), scale_a=asc.unsqueeze(1), scale_b=bsc.unsqueeze(0), out_dtype=torch.bfloat16)
    t = timeit(lambda: torch._scaled_mm(af8, bf8.t(), scale_a=asc.unsqueeze(1), scale_b=bsc.unsqueeze(0), out_dtype=torch.bfloat16))
    print(f"_scaled_mm rowwise 8192^3: {2*M*N*K/t/1e12:.1f} TF/s")
    c_ref = ops.gemm_bt_fp8(aq, asc, bq, bsc)
    rel = ((out.float()-c_ref.float()).norm()/c_ref.float().norm()).item()
    print(f"  vs our fp8 kernel rel={rel:.4f}")
except Exception as e:
    print("_scaled_mm rowwise FAILED:", str(e)[:200])
try:
    t = timeit(lambda: torch._scaled_mm(af8, bf8.t(), scale_a=torch.tensor(1.0, device=dev), scale_b=torch.tensor(1.0, device=dev), out_dtype=torch.bfloat16))
    print(f"_scaled_mm tensorwise 8192^3: {2*M*N*K/t/1e12:.1f} TF/s")
except Exception as e:
    print("_scaled_mm tensorwise FAILED:", str(e)[:200])

# ---- grouped mm ----
for name in ("_grouped_mm", "_scaled_grouped_mm"):
    print(name, "available:", hasattr(torch, name))
E, Tk, H, I = 8, 16384, 4096, 14336
x = torch.randn(Tk, H, dtype=torch.bfloat16, device=dev)
w = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev)
offs = torch.arange(1, E + 1, device=dev, dtype=torch.int32) * (Tk // E)
if hasattr(torch, "_grouped_mm"):
    try:
        out = torch._grouped_mm(x, w.transpose(1, 2), offs=offs)
        t = timeit(lambda: torch._grouped_mm(x, w.transpose(1, 2), offs=offs))
        fl = 2 * Tk * H * I
        print(f"_grouped_mm balanced E8: {fl/t/1e12:.1f} TF/s out {tuple(out.shape)}")
        segs = [0] + [int(o) for o in offs.cpu()]
        xp = torch.zeros(Tk + 128, H, dtype=torch.bfloat16, device=dev); xp[:Tk] = x
        ours = ops.grouped_gemm_bt(xp, w, segs)
        t2 = timeit(lambda: ops.grouped_gemm_bt(xp, w, segs))
        rel = ((out.float() - ours[:Tk].float()).norm() / out.float().norm()).item()
        print(f"ours grouped:            {fl/t2/1e12:.1f} TF/s rel={rel:.4f}")
    except Exception as e:
        print("_grouped_mm FAILED:", str(e)[:300])

# ---- MX block-scaled via _scaled_mm? ----
print("float8_e8m0fnu:", hasattr(torch, "float8_e8m0fnu"))
if hasattr(torch, "float8_e8m0fnu"):
    try:
        amq, ams = ops.quant_mxfp8(a)
        bmq, bms = ops.quant_mxfp8(b)
        sa = ams.view(torch.float8_e8m0fnu)
        sb = bms.view(torch.float8_e8m0fnu)
        out = torch._scaled_mm(amq.view(torch.float8_e4m3fn), bmq.view(torch.float8_e4m3fn).t(),
                               scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16)
        t = timeit(lambda: torch._scaled_mm(amq.view(torch.float8_e4m3fn), bmq.view(torch.float8_e4m3fn).t(),
                                            scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16))
        print(f"_scaled_mm MX 8192^3: {2*M*N*K/t/1e12:.1f} TF/s")
        ours_mx = ops.gemm_bt_mxfp8(amq, ams, bmq, bms)
        rel = ((out.float()-ours_mx.float()).norm()/ours_mx.float().norm()).item()
        print(f"  vs our mx kernel rel={rel:.6f}")
    except Exception as e:
        print("_scaled_mm MX FAILED:", str(e)[:300])
