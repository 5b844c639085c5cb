import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from torch_cgx_amd import _C
def timeit(fn, reps=10):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps * 1e3
for n, lbl in [(256 << 20, "256M(=2^28: generic dequant path)"), ((256 << 20) - 8, "256M-8(fast path)")]:
    x = torch.randn(n, device="cuda", dtype=torch.float16)
    out = torch.empty_like(x)
    comp = _C.quantize(x, 4, 512, True, 0)
    q = timeit(lambda: _C.quantize(x, 4, 512, True, 0))
    d = timeit(lambda: _C.dequantize(comp, out, 4, 512, False))
    print(f"{lbl}: quantize {q:.3f} ms  dequant {d:.3f} ms", flush=True)
