import sys, time, inspect
import torch
sys.path.insert(0, ".")
from senweaver_amd import ops

dev = "cuda:0"
E, Tk, H, I = 8, 16384, 4096, 14336
x = torch.randn(Tk, H, dtype=torch.bfloat16, device=dev)
w = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev)
offs = torch.arange(1, E + 1, device=dev, dtype=torch.int32) * (Tk // E)

xq, xs = ops.quant_fp8(x)          # [Tk,H] u8 + [Tk] f32
wq = torch.empty(E, I, H, dtype=torch.uint8, device=dev)
ws = torch.empty(E, I, dtype=torch.float32, device=dev)
for e in range(E):
    q, s = ops.quant_fp8(w[e])
    wq[e], ws[e] = q, s

def timeit(fn, warm=3, it=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it

try:
    print(torch._scaled_grouped_mm.__doc__ or "no doc")
except Exception as e:
    print("no fn", e)

for desc, call in [
    ("rowwise scales", lambda: torch._scaled_grouped_mm(
        xq.view(torch.float8_e4m3fn), wq.view(torch.float8_e4m3fn).transpose(1, 2),
        xs, ws, offs=offs, out_dtype=torch.bfloat16)),
    ("rowwise unsqueezed", lambda: torch._scaled_grouped_mm(
        xq.view(torch.float8_e4m3fn), wq.view(torch.float8_e4m3fn).transpose(1, 2),
        xs.unsqueeze(1), ws.unsqueeze(2), offs=offs, out_dtype=torch.bfloat16)),
]:
    try:
        out = call()
        t = timeit(call)
        fl = 2 * Tk * H * I
        print(f"{desc}: OK {fl/t/1e12:.1f} TF/s shape {tuple(out.shape)}")
        # numerics vs dequant reference on one segment
        seg = slice(0, Tk // E)
        ref = (xq[seg].view(torch.float8_e4m3fn).float() * xs[seg, None]) @ \
              (wq[0].view(torch.float8_e4m3fn).float() * ws[0][:, None]).t()
        rel = ((out[seg].float() - ref).norm() / ref.norm()).item()
        print(f"  seg0 rel={rel:.4f}")
        break
    except Exception as e:
        print(f"{desc}: FAIL {str(e)[:220]}")
