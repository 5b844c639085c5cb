"""Probe: 2 torchrun ranks sharing ONE GPU through the full cgx backend.

Round-1's version of this probe died with exit code 1 and no traceback; the
root cause was sys.path (script not at the repo root), not RCCL.  With that
fixed, this documents what RCCL actually does with two ranks on one device.
Run:  python -m torch.distributed.run --nnodes=1 --nproc-per-node=2 \
        --master-addr 127.0.0.1 --master-port 29411 tools/stage_tworank.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch_cgx_amd  # noqa: F401

rank = int(os.environ["RANK"])
ws = int(os.environ["WORLD_SIZE"])
torch.cuda.set_device(0)  # both ranks on GPU 0
os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = "4"
dist.init_process_group("cgx", rank=rank, world_size=ws)
t = torch.full((1000,), float(rank + 1), device="cuda:0")
dist.all_reduce(t)
torch.cuda.synchronize()
expected = float(sum(range(1, ws + 1)))
ok = torch.allclose(t, torch.full_like(t, expected), atol=1.0)
print(f"rank {rank}: sum={t[0].item()} expected={expected} ok={ok}",
      flush=True)
dist.destroy_process_group()
