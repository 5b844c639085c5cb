"""Golden (reference-exact) model of the CGX max-min quantizer wire format.

This is the single source of truth for the compressed wire format used by the
whole framework.  The HIP/CDNA4 kernels (csrc/quant_kernels.hip), the C++
engine, and the CPU simulation of the reducers are all tested against this
model, and in deterministic-rounding mode the GPU kernels must match it
byte-for-byte.

Wire format (parity with the reference implementation,
/root/reference/src/common/compression/cuda_compression_operations.cu:68-96,
219-285 and /root/reference/src/common/compressor.cc:401-419):

For a 1-D buffer of ``n`` elements of dtype ``T`` (fp32 / fp16 / bf16),
``bucket_size`` B and ``bits`` q in [1, 8]:

* ``num_buckets = ceil(n / B)``
* meta region: ``2 * num_buckets`` values of type T, interleaved per bucket as
  ``(unit, min)`` where ``unit = (max - min) / (2**q - 1)`` computed in fp32
  and rounded back to T.
* packed region, starting immediately after the meta region: each group of 8
  consecutive values is encoded into a 64-bit word
  ``value = sum(level_j << (j * q))`` and written as its low
  ``q`` bytes, little-endian; total packed bytes ``num_char = ceil(n*q/8)``.
* encode: ``level = min(floor((x - min) * (1/unit) + rand), 2**q - 1)``
  in fp32, with the per-bucket reciprocal ``1/unit`` rounded to fp32 once
  (``level = 0`` when ``unit < EPS``); deterministic rounding uses
  ``rand = 0.5``.
* decode: ``min + unit * level`` computed in dtype T.
* total buffer size: ``2*num_buckets*sizeof(T) + align8(num_char)``
  (+ ``residual*sizeof(T)`` raw tail values when ``skip_incomplete`` and
  ``n % B != 0``).
"""

from __future__ import annotations

import numpy as np
import torch

EPS = 1e-10
PACK_SIZE = 8
ALIGN = 8

_SUPPORTED = (torch.float32, torch.float16, torch.bfloat16)


def align8(x: int) -> int:
    return (x + ALIGN - 1) // ALIGN * ALIGN


def elem_size(dtype: torch.dtype) -> int:
    return torch.tensor([], dtype=dtype).element_size()


def num_buckets(n: int, bucket_size: int) -> int:
    return (n + bucket_size - 1) // bucket_size


def buffer_size(n: int, dtype: torch.dtype, bits: int, bucket_size: int,
                skip_incomplete: bool = False) -> int:
    """Compressed byte size of an n-element slice (reference BufferSize parity)."""
    if n == 0:
        return 0
    es = elem_size(dtype)
    nb = num_buckets(n, bucket_size)
    residuals = 0
    if skip_incomplete:
        nb = n // bucket_size
        residuals = n % bucket_size
        n = nb * bucket_size
    meta = 2 * nb * es
    packed = (n * bits + 7) // 8
    return meta + align8(packed) + residuals * es


def compute_meta(x: torch.Tensor, bits: int, bucket_size: int) -> torch.Tensor:
    """Per-bucket (unit, min) pairs in x.dtype, shape [2*num_buckets]."""
    assert x.dim() == 1 and x.dtype in _SUPPORTED
    n = x.numel()
    nb = num_buckets(n, bucket_size)
    pad = nb * bucket_size - n
    if pad:
        # pad with the first element of the last bucket so max/min are unaffected
        xp = torch.cat([x, x[(nb - 1) * bucket_size].repeat(pad)])
    else:
        xp = x
    xb = xp.view(nb, bucket_size)
    bmax = xb.max(dim=1).values
    bmin = xb.min(dim=1).values
    unit = ((bmax.float() - bmin.float()) / float((1 << bits) - 1)).to(x.dtype)
    meta = torch.empty(2 * nb, dtype=x.dtype)
    meta[0::2] = unit
    meta[1::2] = bmin
    return meta


def encode_levels(x: torch.Tensor, meta: torch.Tensor, bits: int,
                  bucket_size: int, rand=0.5) -> np.ndarray:
    """Quantization levels (uint8 numpy array of len n)."""
    n = x.numel()
    nb = meta.numel() // 2
    unit = meta[0::2].float().numpy()
    bmin = meta[1::2].float().numpy()
    idx = np.arange(n) // bucket_size
    xf = x.float().numpy()
    if isinstance(rand, torch.Tensor):
        rand = rand.float().numpy()
    # encode via the fp32-rounded per-bucket reciprocal (the kernels hoist
    # 1/unit out of the element loop; fp division per element is ~4x VALU
    # cost on CDNA4).  All fp32 ops single-rounded -> bitwise kernel parity.
    with np.errstate(divide="ignore", invalid="ignore"):
        inv_unit = (np.float32(1.0) / unit).astype(np.float32)
        rnd = np.float32(rand) if np.isscalar(rand) else rand.astype(np.float32)
        d = (xf - bmin[idx]).astype(np.float32) * inv_unit[idx] + rnd
    level = np.minimum(np.floor(d), float((1 << bits) - 1))
    level = np.where(unit[idx] < EPS, 0.0, level)
    return level.astype(np.uint8)


def pack_levels(levels: np.ndarray, bits: int) -> np.ndarray:
    """Pack levels into the little-endian groups-of-8 byte stream (uint8)."""
    n = len(levels)
    num_char = (n * bits + 7) // 8
    ngroups = (n + PACK_SIZE - 1) // PACK_SIZE
    lv = np.zeros(ngroups * PACK_SIZE, dtype=np.uint64)
    lv[:n] = levels.astype(np.uint64)
    lv = lv.reshape(ngroups, PACK_SIZE)
    value = np.zeros(ngroups, dtype=np.uint64)
    for j in range(PACK_SIZE):
        value |= lv[:, j] << np.uint64(j * bits)
    by = value[:, None] >> (np.uint64(8) * np.arange(8, dtype=np.uint64))[None, :]
    by = (by & np.uint64(0xFF)).astype(np.uint8)[:, :bits]
    return by.reshape(-1)[:num_char].copy()


def unpack_levels(packed: np.ndarray, n: int, bits: int) -> np.ndarray:
    """Inverse of pack_levels -> uint8 levels of length n."""
    ngroups = (n + PACK_SIZE - 1) // PACK_SIZE
    by = np.zeros((ngroups, 8), dtype=np.uint8)
    take = min(len(packed), ngroups * bits)
    # bytes of group i live at packed[i*bits : i*bits+bits]
    src = packed[:take]
    gi = np.arange(take) // bits
    bi = np.arange(take) % bits
    by[gi, bi] = src
    value = by.view(np.uint64).reshape(ngroups)  # little-endian host assumed
    mask = np.uint64((1 << bits) - 1)
    levels = np.zeros(ngroups * PACK_SIZE, dtype=np.uint8)
    for j in range(PACK_SIZE):
        levels[j::PACK_SIZE] = ((value >> np.uint64(j * bits)) & mask).astype(np.uint8)
    return levels[:n]


def _raw_bytes(t: torch.Tensor) -> torch.Tensor:
    raw = t.view(torch.int16) if t.dtype == torch.bfloat16 else t
    return torch.from_numpy(
        np.frombuffer(raw.contiguous().numpy().tobytes(),
                      dtype=np.uint8).copy())


def _from_raw_bytes(b: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    arr = b.numpy().copy()
    if dtype == torch.float32:
        return torch.from_numpy(arr.view(np.float32).copy())
    return torch.from_numpy(arr.view(np.int16).copy()).view(dtype)


def quantize_ef(x: torch.Tensor, feedback: torch.Tensor, bits: int,
                bucket_size: int, rand=0.5) -> torch.Tensor:
    """Error-feedback quantize: encodes x + feedback and updates feedback in
    place with the new residual (all in fp32, residual rounded to x.dtype).

    The reference shipped EF kernel plumbing but never enabled it
    (cuda_compression_operations.cu:713-725 passes EF=false); this is the
    working version.  Wire format is unchanged."""
    assert feedback.shape == x.shape and feedback.dtype == x.dtype
    xe = (x.float() + feedback.float()).to(x.dtype)
    comp = quantize(xe, bits, bucket_size, rand)
    dec = dequantize(comp, xe.numel(), x.dtype, bits, bucket_size)
    feedback.copy_((xe.float() - dec.float()).to(x.dtype))
    return comp


def quantize(x: torch.Tensor, bits: int, bucket_size: int,
             rand=0.5, skip_incomplete: bool = False) -> torch.Tensor:
    """Compress x -> uint8 buffer of exactly buffer_size(...) bytes.

    Alignment padding bytes are zero-filled (the reference leaves them
    undefined; we define them as zero so byte comparisons are meaningful).
    """
    assert 1 <= bits <= 8
    x = x.contiguous().view(-1)
    n = x.numel()
    if skip_incomplete:
        nb = n // bucket_size
        nq = nb * bucket_size
        r = n - nq
        total = buffer_size(n, x.dtype, bits, bucket_size, True)
        out = torch.zeros(total, dtype=torch.uint8)
        if nq:
            head = quantize(x[:nq], bits, bucket_size, rand)
            out[: head.numel()] = head
        if r:
            rb = _raw_bytes(x[nq:])
            out[total - rb.numel():] = rb
        return out
    total = buffer_size(n, x.dtype, bits, bucket_size)
    out = torch.zeros(total, dtype=torch.uint8)
    meta = compute_meta(x, bits, bucket_size)
    raw = meta.view(torch.int16) if meta.dtype == torch.bfloat16 else meta
    mbytes = torch.from_numpy(
        np.frombuffer(raw.numpy().tobytes(), dtype=np.uint8).copy())
    out[: mbytes.numel()] = mbytes
    levels = encode_levels(x, meta, bits, bucket_size, rand)
    packed = pack_levels(levels, bits)
    off = mbytes.numel()
    out[off: off + len(packed)] = torch.from_numpy(packed)
    return out


def dequantize(buf: torch.Tensor, n: int, dtype: torch.dtype, bits: int,
               bucket_size: int, skip_incomplete: bool = False) -> torch.Tensor:
    """Decompress a quantize() buffer back to an n-element tensor of dtype."""
    buf = buf.contiguous().view(-1)
    if skip_incomplete:
        nb = n // bucket_size
        nq = nb * bucket_size
        r = n - nq
        parts = []
        if nq:
            head_bytes = buffer_size(nq, dtype, bits, bucket_size)
            parts.append(dequantize(buf[:head_bytes], nq, dtype, bits,
                                    bucket_size))
        if r:
            es = elem_size(dtype)
            parts.append(_from_raw_bytes(buf[buf.numel() - r * es:], dtype))
        return torch.cat(parts) if parts else torch.zeros(0, dtype=dtype)
    nb = num_buckets(n, bucket_size)
    es = elem_size(dtype)
    meta_bytes = 2 * nb * es
    meta_np = np.frombuffer(bytes(buf[:meta_bytes].numpy()), dtype=np.uint8)
    if dtype == torch.float32:
        meta = torch.from_numpy(meta_np.view(np.float32).copy())
    else:
        meta = torch.from_numpy(meta_np.view(np.int16).copy()).view(dtype)
    packed = buf[meta_bytes:].numpy()
    levels = unpack_levels(packed, n, bits)
    unit = meta[0::2].to(dtype)
    bmin = meta[1::2].to(dtype)
    idx = torch.from_numpy((np.arange(n) // bucket_size).astype(np.int64))
    lv = torch.from_numpy(levels.copy()).to(dtype)
    # decode arithmetic in dtype T (parity with reference MaxMinDecodeValue)
    return bmin[idx] + unit[idx] * lv
