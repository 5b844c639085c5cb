"""Rank-chunk partitioning for the compressed allreduce.

Python mirror of the C++ engine's partition (csrc/engine.cc), which itself
reproduces the semantics of the reference Quantizer::GetSizesAndOffsets
(/root/reference/src/common/compressor.cc:265-299): a fused chunk of
``num_elements`` values spanning a list of layer slices is split into
``world_size`` contiguous rank-chunks such that

* rank r gets ``remaining / (world_size - r)`` elements (balanced split of
  what is left),
* a split inside a layer is rounded to an alignment unit (8 elements for
  16-bit dtypes, 4 for fp32) so every slice's packed stream starts on a pack
  boundary,
* layer boundaries are preferred split points.

Every rank runs this identically over shared metadata, so compressed chunk
byte sizes can be derived on both sides of a p2p exchange without a size
handshake.
"""

from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

from ..ops import golden


def align_unit_for(dtype: torch.dtype) -> int:
    return 8 if golden.elem_size(dtype) == 2 else 4


def round_to(x: int, unit: int) -> int:
    return (x + unit - 1) // unit * unit


def partition(num_elements: int, world_size: int, global_offset: int,
              layer_numels: Sequence[int], dtype: torch.dtype
              ) -> Tuple[List[int], List[int]]:
    """Return (offsets, sizes) in elements for each rank's chunk."""
    offsets: List[int] = []
    sizes: List[int] = []
    offset = global_offset
    layers = list(layer_numels)
    li = 0
    n_elem = min(layers[0], num_elements) if layers else 0
    unit = align_unit_for(dtype)
    remaining = num_elements
    for rank in range(world_size):
        num_per_node = remaining // (world_size - rank)
        cur = 0
        while cur < num_per_node:
            if n_elem <= num_per_node - cur:
                cur += n_elem
                li += 1
                if li == len(layers):
                    break
                n_elem = min(layers[li], num_elements)
            else:
                aligned = min(round_to(num_per_node - cur, unit), n_elem)
                cur += aligned
                n_elem -= aligned
        remaining -= cur
        sizes.append(cur)
        offsets.append(offset)
        offset += cur
    return offsets, sizes


def layer_slices(layer_numels: Sequence[int], start: int, size: int
                 ) -> List[Tuple[int, int, int]]:
    """Slices of [start, start+size) by layer boundaries.

    Returns a list of (layer_idx, offset_within_chunkspace, slice_numel) where
    offsets are absolute element offsets in the fused-chunk coordinate system
    (layer 0 begins at element 0).
    """
    out: List[Tuple[int, int, int]] = []
    pos = 0
    end = start + size
    for i, n in enumerate(layer_numels):
        lo = max(pos, start)
        hi = min(pos + n, end)
        if hi > lo:
            out.append((i, lo, hi - lo))
        pos += n
        if pos >= end:
            break
    return out


def compressed_chunk_size(layer_numels: Sequence[int],
                          layer_configs: Sequence[Tuple[int, int]],
                          start: int, size: int, dtype: torch.dtype) -> int:
    """Total compressed bytes for a rank chunk [start, start+size).

    layer_configs[i] = (bits, bucket_size) for layer i.
    """
    total = 0
    for (li, off, n) in layer_slices(layer_numels, start, size):
        bits, bucket = layer_configs[li]
        total += golden.buffer_size(n, dtype, bits, bucket)
    return total
