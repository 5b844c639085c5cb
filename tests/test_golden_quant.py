"""Golden wire-format model: self-consistency + parity with the native
buffer-size computation (CPU only)."""

import numpy as np
import pytest
import torch

from torch_cgx_amd.ops import golden

DTYPES = [torch.float32, torch.float16, torch.bfloat16]


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [1, 2, 3, 4, 6, 8])
@pytest.mark.parametrize("n,bucket", [(64, 64), (1000, 512), (1024, 512),
                                      (1025, 512), (131, 1000), (8192, 2048),
                                      (7, 512)])
def test_buffer_size_matches_native(dtype, bits, n, bucket):
    from torch_cgx_amd import _C
    assert golden.buffer_size(n, dtype, bits, bucket) == \
        _C.buffer_size(n, dtype, bits, bucket)


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [2, 4, 8])
def test_roundtrip_error_bound(dtype, bits):
    torch.manual_seed(0)
    for n, bucket in [(512, 512), (1000, 128), (4096, 512), (77, 64)]:
        x = torch.randn(n).to(dtype)
        comp = golden.quantize(x, bits, bucket)
        out = golden.dequantize(comp, n, dtype, bits, bucket)
        # per-bucket error <= unit (deterministic rounding: <= unit/2 + meta
        # rounding in T; unit is a safe bound)
        nb = golden.num_buckets(n, bucket)
        for b in range(nb):
            lo, hi = b * bucket, min((b + 1) * bucket, n)
            xb = x[lo:hi].float()
            unit = (xb.max() - xb.min()) / (2 ** bits - 1)
            err = (xb - out[lo:hi].float()).abs().max()
            tol = unit * 1.05 + 1e-2 * unit + 1e-6
            if dtype != torch.float32:
                tol = tol + (xb.abs().max() * (2 ** -7 if dtype == torch.bfloat16 else 2 ** -10))
            assert err <= tol, (b, err, unit)


@pytest.mark.parametrize("dtype", DTYPES)
def test_constant_input_exact(dtype):
    for n in [1, 7, 64, 1000]:
        x = torch.full((n,), 3.25, dtype=dtype)
        comp = golden.quantize(x, 4, 512)
        out = golden.dequantize(comp, n, dtype, 4, 512)
        assert torch.equal(out, x)


def test_pack_unpack_inverse():
    rng = np.random.default_rng(0)
    for bits in range(1, 9):
        for n in [1, 7, 8, 9, 63, 64, 1000]:
            levels = rng.integers(0, 1 << bits, n).astype(np.uint8)
            packed = golden.pack_levels(levels, bits)
            assert len(packed) == (n * bits + 7) // 8
            back = golden.unpack_levels(packed, n, bits)
            assert np.array_equal(levels, back)


def test_meta_layout_interleaved():
    x = torch.arange(0, 1024, dtype=torch.float32)
    meta = golden.compute_meta(x, 8, 512)
    assert meta.numel() == 4
    # (unit, min) per bucket
    assert meta[1].item() == 0.0
    assert meta[3].item() == 512.0
    assert abs(meta[0].item() - 511 / 255) < 1e-6


def test_stochastic_rand_tensor():
    torch.manual_seed(1)
    x = torch.randn(1024)
    rand = torch.rand(1024)
    meta = golden.compute_meta(x, 4, 512)
    lv_det = golden.encode_levels(x, meta, 4, 512, 0.5)
    lv_sto = golden.encode_levels(x, meta, 4, 512, rand)
    # stochastic levels differ from deterministic by at most 1
    assert (np.abs(lv_det.astype(int) - lv_sto.astype(int)) <= 1).all()


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("n,bucket", [(1000, 512), (1025, 64), (7, 512),
                                      (4096, 512)])
def test_skip_incomplete_roundtrip(dtype, n, bucket):
    torch.manual_seed(n)
    x = torch.randn(n).to(dtype)
    comp = golden.quantize(x, 4, bucket, skip_incomplete=True)
    assert comp.numel() == golden.buffer_size(n, dtype, 4, bucket, True)
    out = golden.dequantize(comp, n, dtype, 4, bucket, skip_incomplete=True)
    r = n % bucket
    if r:
        # raw residual tail is exact
        assert torch.equal(out[-r:], x[-r:])
    nq = n - r
    if nq:
        xb = x[:nq].float().view(-1, bucket)
        unit = (xb.max(1).values - xb.min(1).values) / 15
        err = (xb - out[:nq].float().view(-1, bucket)).abs().max(1).values
        tol = unit * 1.05 + 1e-6 + (0 if dtype == torch.float32 else
                                    xb.abs().max() * 2 ** -7)
        assert (err <= tol).all()


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
def test_error_feedback_time_average_converges(dtype):
    torch.manual_seed(3)
    x = torch.randn(2048).to(dtype)
    fb = torch.zeros_like(x)
    acc = torch.zeros(2048)
    reps = 20
    for k in range(1, reps + 1):
        comp = golden.quantize_ef(x, fb, 2, 512)
        acc += golden.dequantize(comp, x.numel(), dtype, 2, 512).float()
    ef_err = ((acc / reps) - x.float()).abs().max().item()
    one_shot = (golden.dequantize(golden.quantize(x, 2, 512), x.numel(),
                                  dtype, 2, 512).float()
                - x.float()).abs().max().item()
    # EF makes the *time-averaged* transmission unbiased: much closer than a
    # single deterministic quantization
    assert ef_err < one_shot * 0.35, (ef_err, one_shot)
