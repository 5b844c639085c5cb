"""DDP + cgx_hook on CPU (world_size 2): training must match vanilla DDP
(on CPU the backend is gloo passthrough, so results are exact) and the hook
must populate the native layer registry at step 2."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from dist_utils import run_dist


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(32, 64)
        self.fc2 = nn.Linear(64, 64)
        self.fc3 = nn.Linear(64, 10)

    def forward(self, x):
        return self.fc3(torch.relu(self.fc2(torch.relu(self.fc1(x)))))


def _train(model, steps, rank, hook=None, state=None):
    ddp = nn.parallel.DistributedDataParallel(model)
    if hook is not None:
        ddp.register_comm_hook(state, hook)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05)
    torch.manual_seed(1234 + rank)
    for _ in range(steps):
        x = torch.randn(8, 32)
        y = torch.randint(0, 10, (8,))
        loss = nn.functional.cross_entropy(ddp(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in ddp.parameters()]


def _hook_worker(rank, ws):
    import torch_cgx_amd
    from torch_cgx_amd import _C

    torch.manual_seed(7)
    m1 = Net()
    m2 = Net()
    m2.load_state_dict(m1.state_dict())

    _C.clear_registry()
    state = torch_cgx_amd.CGXState(None, layer_min_size=32,
                                   compression_params={"bits": 4,
                                                       "bucket_size": 128})
    p_hooked = _train(m1, 5, rank, torch_cgx_amd.cgx_hook, state)
    p_plain = _train(m2, 5, rank)
    for a, b in zip(p_hooked, p_plain):
        assert torch.allclose(a, b, atol=1e-6), "hooked DDP diverged"
    assert state.step >= 4
    _C.clear_registry()


def test_ddp_hook_cpu_ws2():
    run_dist(_hook_worker, 2)


def _compat_shim_worker(rank, ws):
    # the reference import path must work as a drop-in
    import torch_cgx  # noqa: F401
    from cgx_utils import CGXState, cgx_hook  # noqa: F401
    t = torch.ones(10)
    dist.all_reduce(t)
    assert torch.equal(t, torch.full((10,), float(ws)))


def test_reference_import_compat():
    run_dist(_compat_shim_worker, 2)


def _registry_snapshot_worker(rank, ws):
    import torch_cgx_amd
    from torch_cgx_amd import _C
    _C.clear_registry()
    state = torch_cgx_amd.CGXState(None, layer_min_size=32,
                                   compression_params={"bits": 4,
                                                       "bucket_size": 128})
    _train(Net(), 4, rank, torch_cgx_amd.cgx_hook, state)
    snap = _C.registry_snapshot()
    assert len(snap) >= 1  # DDP registered at least one bucket
    total = sum(sum(numels) for (_idx, numels, _cfgs) in snap)
    n_params = sum(p.numel() for p in Net().parameters())
    assert total == n_params, (total, n_params)
    # layerwise filter: 1-D tensors (biases) registered with bits=32
    saw_fp, saw_q = False, False
    for (_idx, numels, cfgs) in snap:
        for n, (bits, bucket) in zip(numels, cfgs):
            if bits == 32:
                saw_fp = True
            else:
                assert bits == 4 and bucket == 128
                saw_q = True
    assert saw_fp and saw_q
    _C.clear_registry()


def test_registry_snapshot_after_hook():
    run_dist(_registry_snapshot_worker, 2)


def _adaptive_worker(rank, ws):
    import torch_cgx_amd
    from torch_cgx_amd import _C
    from torch_cgx_amd.adaptive import AdaptiveCGXState, adaptive_cgx_hook
    _C.clear_registry()
    state = AdaptiveCGXState(None, schedule=[(0, 8), (4, 2)],
                             layer_min_size=32, bucket_size=128)
    _train(Net(), 8, rank, adaptive_cgx_hook, state)
    snap = _C.registry_snapshot()
    assert snap
    saw2 = False
    for (_i, _n, cfgs) in snap:
        for (bits, _bs) in cfgs:
            assert bits in (2, 32)  # schedule reached 2-bit; fp32 pins stay
            saw2 |= bits == 2
    assert saw2
    _C.clear_registry()


def test_adaptive_schedule():
    run_dist(_adaptive_worker, 2)
