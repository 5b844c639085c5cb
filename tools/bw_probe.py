#!/usr/bin/env python3
"""Memory-bandwidth reference points for the dequant roofline analysis:
write-only (fill_), read+write (copy_), and the fp32 dequant under different
grid caps (CGX_MAX_BLOCKS must be set before the first kernel launch, so
each grid config runs in its own process: tools/bw_probe.sh loops)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torch_cgx_amd import _C

N = 64 << 20


def timeit(fn, reps=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps * 1e3


x = torch.randn(N, device="cuda")
y = torch.empty_like(x)
fill = timeit(lambda: y.fill_(1.0))
copy = timeit(lambda: y.copy_(x))
print(f"fill_ 256MB write-only: {fill:.3f} ms ({0.256 / fill * 1e3:.2f} TB/s)")
print(f"copy_ 256MB r + 256MB w: {copy:.3f} ms ({0.512 / copy * 1e3:.2f} TB/s)")

blocks = os.environ.get("CGX_MAX_BLOCKS", "4096")
for bits in (1, 4, 8):
    comp = _C.quantize(x, bits, 1024, True, 0)
    dt = timeit(lambda: _C.dequantize(comp, y, bits, 1024, False))
    eff = (N * 4 + comp.numel()) / dt / 1e9
    print(f"blocks={blocks} bits={bits}: dequant {dt:.3f} ms ({eff:.2f} TB/s)",
          flush=True)
