#!/usr/bin/env python3
"""Kernel throughput sweep: quantize / dequantize(single + multi-source) for
fp32/fp16/bf16 x bits {1,4,8} at 64M elements, bucket 1024.  Prints ms and
effective TB/s (input+output traffic of each op).  Used to track the
roofline-gap work."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torch_cgx_amd import _C

N = 64 << 20
BUCKET = 1024
REPS = 30


def timeit(fn, reps=REPS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps * 1e3


def run(dtype, bits):
    es = 4 if dtype == torch.float32 else 2
    x = torch.randn(N, dtype=torch.float32).to(dtype).cuda()
    out = torch.empty_like(x)
    comp = _C.quantize(x, bits, BUCKET, True, 0)
    cb = comp.numel()

    qt = timeit(lambda: _C.quantize(x, bits, BUCKET, True, 0))
    q_gb = (N * es + cb) / qt / 1e9  # ms -> bytes/ms = MB/s... careful
    dt = timeit(lambda: _C.dequantize(comp, out, bits, BUCKET, False))
    d_gb = (N * es + cb) / dt / 1e9
    # multi-source decode-accumulate (round-1 shape, 7 peers)
    nsrc = 7
    multi = comp.unsqueeze(0).repeat(nsrc, 1).contiguous()
    mt = timeit(lambda: _C.dequantize_multi(multi, out, bits, BUCKET, True))
    m_gb = (N * es * 2 + nsrc * cb) / mt / 1e9
    name = str(dtype).split('.')[-1]
    print(f"{name:9s} bits={bits}  quantize {qt:7.3f} ms ({q_gb:5.2f} TB/s)  "
          f"dequant {dt:7.3f} ms ({d_gb:5.2f} TB/s)  "
          f"dequant7x {mt:7.3f} ms ({m_gb:5.2f} TB/s)", flush=True)


if __name__ == "__main__":
    for dtype in (torch.float32, torch.float16, torch.bfloat16):
        for bits in (1, 4, 8):
            run(dtype, bits)
