import os, torch, torch.distributed as dist
import torch_cgx_amd
rank = int(os.environ["RANK"]); ws = int(os.environ["WORLD_SIZE"])
torch.cuda.set_device(0)  # both ranks on GPU 0
dist.init_process_group("cgx", rank=rank, world_size=ws)
t = torch.full((1000,), float(rank + 1), device="cuda:0")
os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = "4"
dist.all_reduce(t)
expected = float(sum(range(1, ws + 1)))
ok = torch.allclose(t, torch.full_like(t, expected), atol=1.0)
print(f"rank {rank}: sum={t[0].item()} expected={expected} ok={ok}", flush=True)
dist.destroy_process_group()
