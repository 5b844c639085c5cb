"""Huge-tensor paths on MI355X (288 GB HBM): exercises the 64-bit indexing
branches (`small == false`) that small tensors never reach.

Strategy: bucket-aligned windows quantize independently, so the byte stream
of window [k*B*W : (k+1)*B*W) inside a huge tensor's packed region must
equal the standalone quantization of that window (plus the corresponding
meta slice) — no CPU golden pass over billions of elements needed."""

import pytest
import torch

from torch_cgx_amd.ops import golden

pytestmark = pytest.mark.gpu


def _dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def _check_windows(x, comp, n, dtype, bits, bucket):
    from torch_cgx_amd import _C
    es = golden.elem_size(dtype)
    nb = golden.num_buckets(n, bucket)
    meta_bytes = 2 * nb * es
    # windows of 8 buckets at the start, middle, end
    win_buckets = 8
    win = bucket * win_buckets
    for start_bucket in [0, nb // 2, nb - win_buckets]:
        s = start_bucket * bucket
        e = min(s + win, n)
        piece = x[s:e].contiguous()
        pc = _C.quantize(piece, bits, bucket, False, 0).cpu()
        p_nb = golden.num_buckets(e - s, bucket)
        p_meta = 2 * p_nb * es
        # meta slice equality
        big_meta = comp[2 * start_bucket * es: 2 * start_bucket * es + p_meta]
        assert torch.equal(big_meta, pc[:p_meta]), f"meta @bucket {start_bucket}"
        # packed slice equality (window start is group-aligned: bucket%8==0)
        gofs = (s // 8) * bits
        plen = ((e - s) * bits) // 8
        big_packed = comp[meta_bytes + gofs: meta_bytes + gofs + plen]
        assert torch.equal(big_packed, pc[p_meta: p_meta + plen]), \
            f"packed @bucket {start_bucket}"


@pytest.mark.parametrize("n,dtype,bits,bucket", [
    (300_000_000, torch.float32, 4, 1024),   # 2^28 < n < 2^31: u32 fast path
    (268_435_456, torch.float32, 4, 1024),   # exactly 2^28 (old u32 boundary)
    (2_200_000_000, torch.float16, 4, 1024),  # > 2^31 elements
])
def test_huge_tensor_quantize_windows(n, dtype, bits, bucket):
    from torch_cgx_amd import _C
    torch.manual_seed(1)
    x = torch.empty(n, dtype=dtype, device=_dev())
    x.normal_()
    comp = _C.quantize(x, bits, bucket, False, 0)
    assert comp.numel() == golden.buffer_size(n, dtype, bits, bucket)
    _check_windows(x, comp.cpu(), n, dtype, bits, bucket)

    # decode round-trip error bound on sampled windows
    out = torch.empty_like(x)
    _C.dequantize(comp, out, bits, bucket, False)
    for s in [0, n // 2, n - bucket]:
        seg = slice(s, s + bucket)
        xb = x[seg].float()
        unit = (xb.max() - xb.min()) / ((1 << bits) - 1)
        err = (xb - out[seg].float()).abs().max()
        assert err <= unit * 1.05 + 1e-5, (s, err.item(), unit.item())
    del x, out, comp
    torch.cuda.empty_cache()
