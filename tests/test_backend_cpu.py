"""The cgx backend on CPU tensors (gloo delegation): registration, routing,
collectives, world_size=2 over two processes (runs without any GPU)."""

import os

import pytest
import torch
import torch.distributed as dist

from dist_utils import run_dist


def _allreduce(rank, ws):
    for size in [1, 2, 8, 128, 1024, 100000]:
        for dtype in [torch.float16, torch.float32, torch.int32]:
            t = torch.tensor([rank + 1] * size, dtype=dtype)
            dist.all_reduce(t)
            expected = torch.tensor([(ws * (ws + 1)) // 2] * size, dtype=dtype)
            assert torch.equal(t, expected), (size, dtype)


def test_allreduce_cpu_ws2():
    run_dist(_allreduce, 2)


def _collectives(rank, ws):
    t = torch.full((16,), float(rank))
    dist.broadcast(t, src=0)
    assert torch.equal(t, torch.zeros(16))

    t = torch.full((4,), float(rank + 1))
    out = [torch.zeros(4) for _ in range(ws)]
    dist.all_gather(out, t)
    for p in range(ws):
        assert torch.equal(out[p], torch.full((4,), float(p + 1)))

    dist.barrier()

    r = torch.full((8,), float(rank + 1))
    dist.reduce(r, dst=0)
    if rank == 0:
        assert torch.equal(r, torch.full((8,), float(sum(range(1, ws + 1)))))

    if rank == 0:
        dist.send(torch.arange(5.0), dst=1)
    elif rank == 1:
        rt = torch.zeros(5)
        dist.recv(rt, src=0)
        assert torch.equal(rt, torch.arange(5.0))


def test_collectives_cpu_ws2():
    run_dist(_collectives, 2)


def _min_max(rank, ws):
    t = torch.tensor([float(rank + 1)] * 8)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert torch.equal(t, torch.full((8,), float(ws)))
    t = torch.tensor([float(rank + 1)] * 8)
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    assert torch.equal(t, torch.full((8,), 1.0))


def test_reduce_ops_cpu():
    run_dist(_min_max, 2)


def test_registry_api():
    from torch_cgx_amd import _C
    _C.clear_registry()
    _C.register_layer(0, 0, 1000, 4, 512)
    _C.register_layer(0, 1, 24, 32, 512)
    _C.register_layer(1, 0, 5000, 4, 512)
    _C.set_quantization_bits(0, 0, 8)
    _C.set_quantization_bucket_size(0, 0, 1024)
    _C.clear_registry()


def _ws3(rank, ws):
    t = torch.tensor([rank + 1.0] * 33)
    dist.all_reduce(t)
    assert torch.equal(t, torch.full((33,), 6.0))


def test_allreduce_cpu_ws3():
    run_dist(_ws3, 3)


def _subgroup(rank, ws):
    g = dist.new_group([0, 1])
    t = torch.full((8,), float(rank + 1))
    if rank in (0, 1):
        dist.all_reduce(t, group=g)
        assert torch.equal(t, torch.full((8,), 3.0))
    dist.barrier()


def test_new_group_cpu():
    run_dist(_subgroup, 2)


def _more_collectives(rank, ws):
    # gather
    t = torch.full((4,), float(rank + 1))
    out = [torch.zeros(4) for _ in range(ws)] if rank == 0 else None
    dist.gather(t, out, dst=0)
    if rank == 0:
        for p in range(ws):
            assert torch.equal(out[p], torch.full((4,), float(p + 1)))
    # scatter
    r = torch.zeros(4)
    src = [torch.full((4,), float(p + 10)) for p in range(ws)] \
        if rank == 0 else None
    dist.scatter(r, src, src=0)
    assert torch.equal(r, torch.full((4,), float(rank + 10)))
    # alltoall
    ins = list(torch.arange(float(ws * 2)).add(rank * 100).chunk(ws))
    outs = list(torch.zeros(ws * 2).chunk(ws))
    dist.all_to_all(outs, ins)
    for p in range(ws):
        assert torch.equal(outs[p], torch.arange(float(ws * 2))
                           .add(p * 100).chunk(ws)[rank])


def test_gather_scatter_alltoall_cpu():
    run_dist(_more_collectives, 2)


def _base_variants(rank, ws):
    # all_gather_into_tensor (_allgather_base delegation)
    t = torch.full((4,), float(rank + 1))
    out = torch.zeros(4 * ws)
    dist.all_gather_into_tensor(out, t)
    for p in range(ws):
        assert torch.equal(out[p * 4:(p + 1) * 4],
                           torch.full((4,), float(p + 1)))
    # alltoall_base equal splits
    inp = torch.arange(float(ws * 3)) + rank * 100
    outp = torch.zeros(ws * 3)
    dist.all_to_all_single(outp, inp)
    for p in range(ws):
        expect = torch.arange(float(ws * 3))[rank * 3:(rank + 1) * 3] + p * 100
        assert torch.equal(outp[p * 3:(p + 1) * 3], expect)


def test_base_collectives_cpu():
    run_dist(_base_variants, 2)


def _subgroup_worker(rank, ws):
    # dist.new_group on the cgx backend (the reference exposed group names
    # "for FSDP"; here subgroups get their own prefixed store + delegate)
    import torch
    g = dist.new_group([0, 1])
    t = torch.full((10,), float(rank + 1))
    if rank in (0, 1):
        dist.all_reduce(t, group=g)
        assert torch.equal(t, torch.full((10,), 3.0)), t
    else:
        assert torch.equal(t, torch.full((10,), float(rank + 1)))
    dist.barrier()


def test_subgroup_cpu_ws3():
    from dist_utils import run_dist
    run_dist(_subgroup_worker, 3)


def _anysource_worker(rank, ws):
    import torch
    t = torch.zeros(4)
    if rank == 1:
        dist.send(torch.full((4,), 7.0), dst=0)
    elif rank == 0:
        src = dist.recv(t, src=None)  # recvAnysource via the gloo delegate
        assert src == 1, src
        assert torch.equal(t, torch.full((4,), 7.0))
    dist.barrier()


def test_recv_anysource_cpu():
    from dist_utils import run_dist
    run_dist(_anysource_worker, 2)

