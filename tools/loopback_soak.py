"""Robustness soak: many loopback allreduces at varying world sizes, shapes,
bits and reduction modes; verifies the ws-invariant (all ranks bitwise
identical) every iteration.  Exercises the hub rendezvous, staging reuse,
descriptor ring and both reducers back-to-back."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torch_cgx_amd import _C

torch.manual_seed(0)
it = 0
for rep in range(int(os.environ.get("SOAK_REPS", "4"))):
    for ws in (2, 3, 4, 8):
        for bits in (1, 4, 8):
            for n in (1000, 65536, 1_000_003):
                os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = str(bits)
                os.environ["CGX_COMPRESSION_BUCKET_SIZE"] = "1024"
                os.environ["CGX_INNER_REDUCTION_TYPE"] = (
                    "Ring" if (it % 3 == 2 and ws > 2) else "SRA")
                bufs = [torch.randn(n, device="cuda") for _ in range(ws)]
                _C.loopback_allreduce(bufs)
                for b in bufs[1:]:
                    assert torch.equal(b, bufs[0]), (ws, bits, n, it)
                it += 1
print(f"soak ok: {it} allreduces", flush=True)
