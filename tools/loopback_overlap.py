"""Overlap evidence harness.

Runs a multi-chunk loopback allreduce at ws=2 on one GPU so rocprofv3 can
record the timeline: with the engine's three-stream pipeline, chunk k+1's
quantize kernels (on the caller stream) must overlap chunk k's transport
copies / decode work (comm/deq streams).  Under RCCL the copies are xGMI
p2p; under loopback they are D2D SDMA copies — the stream/event schedule
being exercised is identical (csrc/engine.cc sra_chunk).

Run under: rocprofv3 --kernel-trace --memory-copy-trace --stats -d OUT -- \
    python tools/loopback_overlap.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torch_cgx_amd import _C

os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = "4"
os.environ["CGX_COMPRESSION_BUCKET_SIZE"] = "1024"
os.environ["CGX_FUSION_BUFFER_SIZE_MB"] = "16"  # 64M fp32 -> 16 chunks
os.environ.setdefault("CGX_TIMINGS", "1")

n = 64 * 1024 * 1024
ws = 2
torch.manual_seed(0)
base = [torch.randn(n, device="cuda:0") for _ in range(ws)]
for it in range(5):
    bufs = [b.clone() for b in base]
    _C.loopback_allreduce(bufs)
torch.cuda.synchronize()
print("overlap harness done:", (bufs[0][:4]).tolist(), flush=True)
