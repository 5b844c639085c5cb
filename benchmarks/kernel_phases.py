#!/usr/bin/env python3
"""Phase breakdown of the quantize path: fused (bucket%8==0) vs two-pass
(meta kernel + generic pack kernel, bucket%8!=0), per dtype.  Run under
rocprofv3 --stats to get the per-kernel split."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch_cgx_amd import _C

n = 64 << 20
for dtype in (torch.float32, torch.float16):
    x = torch.randn(n, dtype=torch.float32).to(dtype).cuda()
    out = torch.empty_like(x)
    for bucket in (1024, 1000):
        for _ in range(3):
            c = _C.quantize(x, 4, bucket, True, 0)
        torch.cuda.synchronize()
        t = time.perf_counter()
        for _ in range(20):
            c = _C.quantize(x, 4, bucket, True, 0)
        torch.cuda.synchronize()
        q = (time.perf_counter() - t) / 20 * 1e3
        for _ in range(3):
            _C.dequantize(c, out, 4, bucket, False)
        torch.cuda.synchronize()
        t = time.perf_counter()
        for _ in range(20):
            _C.dequantize(c, out, 4, bucket, False)
        torch.cuda.synchronize()
        d = (time.perf_counter() - t) / 20 * 1e3
        print(f"{dtype} bucket={bucket}: quantize {q:.3f} ms  dequant {d:.3f} ms",
              flush=True)
