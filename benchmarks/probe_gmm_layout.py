import sys, time
import torch
sys.path.insert(0, ".")
from senweaver_amd import ops

dev = "cuda:0"
E, Tk, H, I = 8, 8192, 4096, 14336
x = torch.randn(Tk, H, dtype=torch.bfloat16, device=dev)
w = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev)      # [E,N,K]
wt = w.transpose(1, 2).contiguous()                              # [E,K,N]
offs = torch.arange(1, E + 1, device=dev, dtype=torch.int32) * (Tk // E)

def timeit(fn, warm=3, it=15):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it

t_view = timeit(lambda: torch._grouped_mm(x, w.transpose(1, 2), offs=offs))
t_cont = timeit(lambda: torch._grouped_mm(x, wt, offs=offs))
fl = 2 * Tk * H * I
o1 = torch._grouped_mm(x, w.transpose(1, 2), offs=offs)
o2 = torch._grouped_mm(x, wt, offs=offs)
print(f"view [E,N,K].t: {fl/t_view/1e12:7.1f} TF/s   contiguous [E,K,N]: {fl/t_cont/1e12:7.1f} TF/s   equal={torch.equal(o1, o2)}")
