"""CPU simulation of the compressed Scatter-Reduce-AllGather allreduce.

This mirrors, step for step, the algorithm the C++ engine runs over RCCL
(csrc/engine.cc), in pure Python with the golden quantizer.  It exists to

* validate the algorithm (error bound, equal-input exactness) on CPU with no
  GPU or process group, for any world size,
* provide the expected byte streams / results that the single-device GPU
  simulation test compares against.

Algorithm (parity with the reference scatter_reduce_allgather.cc:94-202):

round 1  each rank r compresses, for every peer k != r, its copy of chunk k
         and sends it to k; rank k decompress-ACCUMULATES every incoming
         version of chunk k into its own (raw) copy.
self-q   rank k compresses the reduced chunk k once; the compressed bytes are
         both decompressed locally (overwriting chunk k) and sent in round 2,
         so every rank ends with bit-identical values.
round 2  rank k sends the compressed reduced chunk k to all peers, which
         decompress-overwrite their copy of chunk k.
"""

from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

from ..ops import golden
from . import partition as P


def compress_chunk(buf: torch.Tensor, layer_numels: Sequence[int],
                   layer_configs: Sequence[Tuple[int, int]],
                   start: int, size: int, rand=0.5,
                   skip: bool = False) -> torch.Tensor:
    """Compress [start, start+size) of buf slice-by-slice -> uint8 tensor."""
    parts = []
    for (li, off, n) in P.layer_slices(layer_numels, start, size):
        bits, bucket = layer_configs[li]
        parts.append(golden.quantize(buf[off: off + n], bits, bucket, rand,
                                     skip_incomplete=skip))
    if not parts:
        return torch.zeros(0, dtype=torch.uint8)
    return torch.cat(parts)


def decompress_chunk(comp: torch.Tensor, buf: torch.Tensor,
                     layer_numels: Sequence[int],
                     layer_configs: Sequence[Tuple[int, int]],
                     start: int, size: int, add: bool,
                     skip: bool = False) -> None:
    """Decompress a compress_chunk() stream into buf (overwrite or +=)."""
    pos = 0
    for (li, off, n) in P.layer_slices(layer_numels, start, size):
        bits, bucket = layer_configs[li]
        nbytes = golden.buffer_size(n, buf.dtype, bits, bucket,
                                    skip_incomplete=skip)
        vals = golden.dequantize(comp[pos: pos + nbytes], n, buf.dtype, bits,
                                 bucket, skip_incomplete=skip)
        if add:
            buf[off: off + n] += vals
        else:
            buf[off: off + n] = vals
        pos += nbytes


def sra_allreduce(tensors: List[torch.Tensor],
                  layer_numels: Sequence[int],
                  layer_configs: Sequence[Tuple[int, int]],
                  rand=0.5, skip: bool = False) -> List[torch.Tensor]:
    """Simulate the compressed SRA over len(tensors) ranks; returns results."""
    ws = len(tensors)
    dtype = tensors[0].dtype
    n = tensors[0].numel()
    assert sum(layer_numels) == n
    offsets, sizes = P.partition(n, ws, 0, layer_numels, dtype)
    out = [t.clone().view(-1) for t in tensors]

    # round 1: scatter-reduce
    for k in range(ws):
        for p in range(ws):
            if p == k:
                continue
            comp = compress_chunk(out[p], layer_numels, layer_configs,
                                  offsets[k], sizes[k], rand, skip)
            decompress_chunk(comp, out[k], layer_numels, layer_configs,
                             offsets[k], sizes[k], add=True, skip=skip)

    # self-quantize + round 2: allgather
    for k in range(ws):
        comp = compress_chunk(out[k], layer_numels, layer_configs,
                              offsets[k], sizes[k], rand, skip)
        for p in range(ws):
            decompress_chunk(comp, out[p], layer_numels, layer_configs,
                             offsets[k], sizes[k], add=False, skip=skip)

    return [o.view(tensors[i].shape) for i, o in enumerate(out)]


def ring_allreduce(tensors: List[torch.Tensor],
                   layer_numels: Sequence[int],
                   layer_configs: Sequence[Tuple[int, int]],
                   rand=0.5) -> List[torch.Tensor]:
    """Simulate the compressed Ring allreduce (csrc engine ring_chunk parity;
    reference ring.cc:139-226): ws-1 reduce-scatter hops with per-hop
    requantize of the running partial sum, then allgather hops forwarding the
    once-quantized reduced segments."""
    ws = len(tensors)
    dtype = tensors[0].dtype
    n = tensors[0].numel()
    offsets, sizes = P.partition(n, ws, 0, layer_numels, dtype)
    out = [t.clone().view(-1) for t in tensors]

    # reduce-scatter: rank r sends chunk (r-s)%ws to r+1, accumulates
    # chunk (r-s-1)%ws received from r-1
    for s in range(ws - 1):
        comps = {}
        for r in range(ws):
            sc = (r - s) % ws
            comps[r] = compress_chunk(out[r], layer_numels, layer_configs,
                                      offsets[sc], sizes[sc], rand)
        for r in range(ws):
            rc = (r - s - 1) % ws
            decompress_chunk(comps[(r - 1) % ws], out[r], layer_numels,
                             layer_configs, offsets[rc], sizes[rc], add=True)

    # allgather: chunk k's fully-reduced segment lives on rank (k-1)%ws;
    # its quantized bytes are forwarded unchanged and decoded everywhere
    for k in range(ws):
        producer = (k - 1) % ws
        comp = compress_chunk(out[producer], layer_numels, layer_configs,
                              offsets[k], sizes[k], rand)
        for r in range(ws):
            decompress_chunk(comp, out[r], layer_numels, layer_configs,
                             offsets[k], sizes[k], add=False)

    return [o.view(tensors[i].shape) for i, o in enumerate(out)]
