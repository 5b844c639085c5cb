"""Kernel times at DDP-bucket scale (25MB fp32 = 6.5M elems, bucket 1024,
4-bit) — the shapes the 8-GPU bench actually runs per bucket."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch_cgx_amd import _C

def timeit(fn, reps=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps * 1e6  # us

n = 6_553_600  # 25 MB fp32
x = torch.randn(n, device="cuda")
out = torch.empty_like(x)
comp = _C.quantize(x, 4, 1024, True, 0)
q = timeit(lambda: _C.quantize(x, 4, 1024, True, 0))
d = timeit(lambda: _C.dequantize(comp, out, 4, 1024, False))
ch = n // 8
xc = torch.randn(ch, device="cuda")
compc = _C.quantize(xc, 4, 1024, True, 0)
multi = compc.unsqueeze(0).repeat(7, 1).contiguous()
outc = torch.empty_like(xc)
m = timeit(lambda: _C.dequantize_multi(multi, outc, 4, 1024, True))
print(f"25MB bucket: quantize {q:.1f} us  dequant {d:.1f} us  "
      f"7-src decode of 1/8 chunk {m:.1f} us", flush=True)
