import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch_cgx_amd import _C
n = 64 << 20
dt = torch.float16 if os.environ.get("QDT") == "f16" else torch.float32
x = torch.randn(n, dtype=torch.float32).to(dt).cuda()
for _ in range(20):
    c = _C.quantize(x, 4, 1024, True, 0)
torch.cuda.synchronize()
print("done")
