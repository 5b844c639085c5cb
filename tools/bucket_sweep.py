import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch_cgx_amd import _C
N = 64 << 20
def timeit(fn, reps=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps * 1e3
for dt in (torch.float32, torch.float16):
    es = 4 if dt == torch.float32 else 2
    x = torch.randn(N, dtype=torch.float32).to(dt).cuda()
    out = torch.empty_like(x)
    for bucket in (128, 256, 512, 1024, 2048):
        comp = _C.quantize(x, 4, bucket, True, 0)
        q = timeit(lambda: _C.quantize(x, 4, bucket, True, 0))
        d = timeit(lambda: _C.dequantize(comp, out, 4, bucket, False))
        print(f"{str(dt).split('.')[-1]:8s} bucket={bucket:5d}: "
              f"quantize {q:.3f} ms ({(N*es+comp.numel())/q/1e9:.2f} TB/s)  "
              f"dequant {d:.3f} ms", flush=True)
