"""Rank-chunk partition: native C++ walk == Python mirror, plus invariants."""

import random

import pytest
import torch

from torch_cgx_amd.parallel import partition as P


def cases():
    rng = random.Random(42)
    out = []
    for _ in range(50):
        nl = rng.randint(1, 8)
        layers = [rng.randint(1, 5000) for _ in range(nl)]
        ws = rng.choice([1, 2, 3, 4, 7, 8])
        dtype = rng.choice([torch.float32, torch.float16])
        out.append((sum(layers), ws, layers, dtype))
    out.append((64 * 1024 * 1024, 8, [64 * 1024 * 1024], torch.float16))
    out.append((5, 8, [5], torch.float32))
    out.append((1, 2, [1], torch.float32))
    return out


@pytest.mark.parametrize("n,ws,layers,dtype", cases())
def test_native_matches_python(n, ws, layers, dtype):
    from torch_cgx_amd import _C
    offs_py, sizes_py = P.partition(n, ws, 0, layers, dtype)
    es = 2 if dtype in (torch.float16, torch.bfloat16) else 4
    offs_c, sizes_c = _C.partition(n, ws, layers, es)
    assert offs_py == list(offs_c)
    assert sizes_py == list(sizes_c)


@pytest.mark.parametrize("n,ws,layers,dtype", cases())
def test_invariants(n, ws, layers, dtype):
    offs, sizes = P.partition(n, ws, 0, layers, dtype)
    assert len(offs) == ws and len(sizes) == ws
    assert sum(sizes) == n
    # contiguous coverage
    pos = 0
    for o, s in zip(offs, sizes):
        assert o == pos
        pos += s


def test_layer_slices_cover():
    layers = [100, 50, 333]
    offs, sizes = P.partition(sum(layers), 4, 0, layers, torch.float32)
    seen = []
    for o, s in zip(offs, sizes):
        for (li, off, nn) in P.layer_slices(layers, o, s):
            seen.append((off, nn))
    seen.sort()
    pos = 0
    for off, nn in seen:
        assert off == pos
        pos += nn
    assert pos == sum(layers)


def test_compressed_chunk_size():
    layers = [1000, 2048]
    cfgs = [(4, 512), (8, 64)]
    offs, sizes = P.partition(sum(layers), 2, 0, layers, torch.float32)
    total = sum(
        P.compressed_chunk_size(layers, cfgs, o, s, torch.float32)
        for o, s in zip(offs, sizes))
    from torch_cgx_amd.ops import golden
    # whole-layer compression equals the sum over rank chunks only when no
    # rank boundary splits a layer; just sanity-check positivity + alignment
    assert total > 0 and total % 1 == 0
    assert golden.buffer_size(1000, torch.float32, 4, 512) > 0
