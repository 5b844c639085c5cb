"""Training-convergence evidence on CPU: a DDP comm hook that routes
gradients through the golden compressed-SRA simulation must train a small
model to (approximately) the same loss as exact fp32 averaging."""

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from dist_utils import run_dist


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.net = nn.Sequential(nn.Linear(16, 64), nn.ReLU(),
                                 nn.Linear(64, 64), nn.ReLU(),
                                 nn.Linear(64, 4))

    def forward(self, x):
        return self.net(x)


def _compressed_hook(state, bucket):
    """Allgather the bucket, run the compressed SRA simulation locally, and
    return the (identical on every rank) reduced result."""
    from torch_cgx_amd.parallel import sra_sim
    ws = dist.get_world_size()
    buf = bucket.buffer()
    gathered = [torch.zeros_like(buf) for _ in range(ws)]
    dist.all_gather(gathered, buf)
    bits, bucket_size = state
    out = sra_sim.sra_allreduce(gathered, [buf.numel()],
                                [(bits, bucket_size)])
    fut = torch.futures.Future()
    fut.set_result(out[0] / ws)
    return fut


def _train(rank, steps, hook_state=None):
    torch.manual_seed(42)  # identical init on all ranks
    model = nn.parallel.DistributedDataParallel(Net())
    if hook_state is not None:
        model.register_comm_hook(hook_state, _compressed_hook)
    opt = torch.optim.SGD(model.parameters(), lr=0.2, momentum=0.9)
    # fixed per-rank shards of a memorizable dataset
    torch.manual_seed(100 + rank)
    nb = 4
    data = torch.randn(nb, 32, 16)
    torch.manual_seed(7)
    w = torch.randn(16, 4)
    targets = (data @ w).argmax(-1)
    losses = []
    for s in range(steps):
        b = s % nb
        loss = nn.functional.cross_entropy(model(data[b]), targets[b])
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _worker(rank, ws):
    steps = 120
    exact = _train(rank, steps)
    comp8 = _train(rank, steps, hook_state=(8, 128))
    comp4 = _train(rank, steps, hook_state=(4, 128))
    e_tail = sum(exact[-10:]) / 10
    c8_tail = sum(comp8[-10:]) / 10
    c4_tail = sum(comp4[-10:]) / 10
    assert e_tail < exact[0] * 0.3, "exact training did not converge"
    # 8-bit compression tracks exact training closely; 4-bit still converges
    assert c8_tail < exact[0] * 0.35, (c8_tail, exact[0], e_tail)
    assert c4_tail < exact[0] * 0.6, (c4_tail, exact[0], e_tail)


def test_compressed_training_converges():
    run_dist(_worker, 2)
