"""CDNA4 kernel numerics on a real MI355X: byte-exact parity with the golden
wire-format model (deterministic rounding) and decode parity."""

import numpy as np
import pytest
import torch

from torch_cgx_amd.ops import golden

pytestmark = pytest.mark.gpu

DTYPES = [torch.float32, torch.float16, torch.bfloat16]
CASES = [  # (n, bucket)
    (512, 512), (1024, 512), (1000, 512), (64, 64), (4096, 1024),
    (1 << 20, 512), (1025, 2048), (7, 512), (131, 64),
    (1000, 1000),      # bucket % 8 != 0 -> generic path
    (5000, 120),       # bucket % 8 == 0 small
    (9, 7),            # tiny, bucket % 8 != 0
    (100000, 8192),    # bucket > register stash capacity
]


def _dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [1, 2, 3, 4, 6, 8])
@pytest.mark.parametrize("n,bucket", CASES)
def test_quantize_matches_golden_bytes(dtype, bits, n, bucket):
    from torch_cgx_amd import _C
    torch.manual_seed(n * bits)
    x = torch.randn(n).to(dtype)
    xg = x.to(_dev())
    comp_gpu = _C.quantize(xg, bits, bucket, False, 0).cpu()
    comp_gold = golden.quantize(x, bits, bucket)
    assert comp_gpu.numel() == comp_gold.numel()
    if not torch.equal(comp_gpu, comp_gold):
        diff = (comp_gpu != comp_gold).nonzero().flatten()
        raise AssertionError(
            f"byte mismatch at {diff[:10].tolist()} of {comp_gpu.numel()} "
            f"(n={n} bucket={bucket} bits={bits} dtype={dtype})")


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [2, 4, 8])
@pytest.mark.parametrize("n,bucket", CASES)
def test_dequantize_matches_golden(dtype, bits, n, bucket):
    from torch_cgx_amd import _C
    torch.manual_seed(n * bits + 1)
    x = torch.randn(n).to(dtype)
    comp = golden.quantize(x, bits, bucket)
    expected = golden.dequantize(comp, n, dtype, bits, bucket)
    out = torch.empty(n, dtype=dtype, device=_dev())
    _C.dequantize(comp.to(_dev()), out, bits, bucket, False)
    assert torch.equal(out.cpu(), expected)


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
def test_dequantize_add_accumulates_in_T(dtype):
    from torch_cgx_amd import _C
    torch.manual_seed(3)
    n, bits, bucket = 4096, 4, 512
    x = torch.randn(n).to(dtype)
    base = torch.randn(n).to(dtype)
    comp = golden.quantize(x, bits, bucket)
    dec = golden.dequantize(comp, n, dtype, bits, bucket)
    expected = (base.float() + dec.float()).to(dtype) if dtype != torch.float32 \
        else base + dec
    out = base.to(_dev()).clone()
    _C.dequantize(comp.to(_dev()), out, bits, bucket, True)
    assert torch.equal(out.cpu(), expected)


@pytest.mark.parametrize("nsrc", [1, 3, 7])
def test_dequantize_multi_matches_sequential(nsrc):
    from torch_cgx_amd import _C
    torch.manual_seed(nsrc)
    n, bits, bucket, dtype = 8192, 4, 512, torch.float32
    comps = []
    for s in range(nsrc):
        x = torch.randn(n)
        comps.append(golden.quantize(x, bits, bucket))
    stride = comps[0].numel()
    stacked = torch.stack(comps).to(_dev())
    base = torch.randn(n)

    # sequential T-precision accumulation on GPU (reference semantics)
    seq = base.to(_dev()).clone()
    for s in range(nsrc):
        _C.dequantize(stacked[s].contiguous(), seq, bits, bucket, True)

    multi = base.to(_dev()).clone()
    _C.dequantize_multi(stacked, multi, bits, bucket, True)
    assert torch.equal(multi.cpu(), seq.cpu())


def test_stochastic_rounding_statistics():
    from torch_cgx_amd import _C
    torch.manual_seed(5)
    n, bits, bucket = 1 << 20, 4, 512
    x = torch.randn(n, device=_dev())
    outs = torch.zeros(n, device=_dev())
    reps = 16
    for seed in range(reps):
        comp = _C.quantize(x, bits, bucket, True, seed)
        out = torch.empty_like(x)
        _C.dequantize(comp, out, bits, bucket, False)
        # each sample within one quantization unit of x
        unit = (x.view(-1, bucket).max(1).values -
                x.view(-1, bucket).min(1).values).repeat_interleave(bucket) / 15
        assert ((out - x).abs() <= unit + 1e-6).all()
        outs += out
    mean_err = ((outs / reps) - x).abs().mean().item()
    det = _C.quantize(x, bits, bucket, False, 0)
    dout = torch.empty_like(x)
    _C.dequantize(det, dout, bits, bucket, False)
    det_err = (dout - x).abs().mean().item()
    # stochastic mean converges toward x: noticeably below a single
    # deterministic rounding's mean error
    assert mean_err < det_err * 0.7, (mean_err, det_err)


def _best_of(measure, rounds=2):
    """Floor guards: take the best of N measurement rounds — a rare
    environmental hiccup (one slow round observed ~1 in 7 fresh-box suite
    runs, 20x off) must not fail the suite, a real regression still does."""
    best = 0.0
    for _ in range(rounds):
        torch.cuda.synchronize()
        best = max(best, measure())
    return best


def test_quantize_throughput_floor():
    """Regression guard: 64M fp32 4-bit quantize must stream at >= 1.8 TB/s
    effective input bandwidth on MI355X (round-2 measured 2.4-2.6 TB/s;
    write-only roofline ~6.5 TB/s)."""
    from torch_cgx_amd import _C
    n, bits, bucket = 64 << 20, 4, 1024
    x = torch.randn(n, device=_dev())
    import time

    def measure():
        for _ in range(3):
            _C.quantize(x, bits, bucket, True, 1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 10
        for _ in range(reps):
            _C.quantize(x, bits, bucket, True, 1)
        torch.cuda.synchronize()
        return n * 4 / ((time.perf_counter() - t0) / reps) / 1e9

    gbps = _best_of(measure)
    print(f"quantize 64M fp32 4-bit: {gbps:.0f} GB/s input")
    assert gbps > 1800, f"quantize too slow: {gbps:.0f} GB/s"


def test_dequantize_throughput_floor():
    from torch_cgx_amd import _C
    n, bits, bucket = 64 << 20, 4, 1024
    x = torch.randn(n, device=_dev())
    comp = _C.quantize(x, bits, bucket, False, 0)
    out = torch.empty_like(x)
    import time

    def measure():
        for _ in range(3):
            _C.dequantize(comp, out, bits, bucket, False)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 10
        for _ in range(reps):
            _C.dequantize(comp, out, bits, bucket, False)
        torch.cuda.synchronize()
        return n * 4 / ((time.perf_counter() - t0) / reps) / 1e9

    gbps = _best_of(measure)
    print(f"dequantize 64M fp32 4-bit: {gbps:.0f} GB/s out")
    # round-2 branch-free fast path measured ~3.0 TB/s output rate
    assert gbps > 2400, f"dequantize too slow: {gbps:.0f} GB/s"


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [2, 4, 8])
@pytest.mark.parametrize("n,bucket", [(1000, 512), (1025, 64), (7, 512),
                                      (4096, 512), (131, 8)])
def test_skip_incomplete_matches_golden(dtype, bits, n, bucket):
    from torch_cgx_amd import _C
    torch.manual_seed(n + bits)
    x = torch.randn(n).to(dtype)
    comp_gold = golden.quantize(x, bits, bucket, skip_incomplete=True)
    comp_gpu = _C.quantize(x.to(_dev()), bits, bucket, False, 0, True).cpu()
    assert torch.equal(comp_gpu, comp_gold)
    expected = golden.dequantize(comp_gold, n, dtype, bits, bucket,
                                 skip_incomplete=True)
    out = torch.empty(n, dtype=dtype, device=_dev())
    _C.dequantize(comp_gold.to(_dev()), out, bits, bucket, False, True)
    assert torch.equal(out.cpu(), expected)


def test_skip_incomplete_dequant_add():
    from torch_cgx_amd import _C
    torch.manual_seed(9)
    n, bits, bucket = 1025, 4, 64
    x = torch.randn(n)
    base = torch.randn(n)
    comp = golden.quantize(x, bits, bucket, skip_incomplete=True)
    dec = golden.dequantize(comp, n, torch.float32, bits, bucket,
                            skip_incomplete=True)
    expected = base + dec
    out = base.to(_dev()).clone()
    _C.dequantize(comp.to(_dev()), out, bits, bucket, True, True)
    assert torch.equal(out.cpu(), expected)


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("bits", [2, 4, 8])
@pytest.mark.parametrize("n,bucket", [(4096, 512), (1000, 512), (131, 64),
                                      (4096, 1024)])
def test_error_feedback_matches_golden(dtype, bits, n, bucket):
    from torch_cgx_amd import _C
    torch.manual_seed(n + bits)
    x = torch.randn(n).to(dtype)
    fb_gold = (torch.randn(n) * 0.1).to(dtype)
    fb_gpu = fb_gold.clone().to(_dev())
    comp_gold = golden.quantize_ef(x.clone(), fb_gold, bits, bucket)
    comp_gpu = _C.quantize(x.to(_dev()), bits, bucket, False, 0, False,
                           fb_gpu).cpu()
    assert torch.equal(comp_gpu, comp_gold)
    assert torch.equal(fb_gpu.cpu(), fb_gold)


def test_error_feedback_repeated_use():
    from torch_cgx_amd import _C
    torch.manual_seed(11)
    n, bits, bucket = 1 << 16, 2, 512
    x = torch.randn(n, device=_dev())
    fb = torch.zeros_like(x)
    acc = torch.zeros_like(x)
    reps = 16
    out = torch.empty_like(x)
    for _ in range(reps):
        comp = _C.quantize(x, bits, bucket, False, 0, False, fb)
        _C.dequantize(comp, out, bits, bucket, False)
        acc += out
    ef_err = ((acc / reps) - x).abs().max().item()
    one = _C.quantize(x, bits, bucket, False, 0)
    _C.dequantize(one, out, bits, bucket, False)
    one_err = (out - x).abs().max().item()
    assert ef_err < one_err * 0.35, (ef_err, one_err)


def test_multisource_dequant_throughput_floor():
    """Regression guard for the SRA round-1 decode: 7-source accumulate of
    64M fp32 4-bit must stream at >= 2.5 TB/s effective (round-2 measured
    ~4 TB/s)."""
    from torch_cgx_amd import _C
    n, bits, bucket, nsrc = 64 << 20, 4, 1024, 7
    x = torch.randn(n, device=_dev())
    comp = _C.quantize(x, bits, bucket, True, 0)
    multi = comp.unsqueeze(0).repeat(nsrc, 1).contiguous()
    out = torch.empty_like(x)
    import time

    def measure():
        for _ in range(3):
            _C.dequantize_multi(multi, out, bits, bucket, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 10
        for _ in range(reps):
            _C.dequantize_multi(multi, out, bits, bucket, True)
        torch.cuda.synchronize()
        return (n * 4 * 2 + nsrc * comp.numel()) /             ((time.perf_counter() - t0) / reps) / 1e9

    eff = _best_of(measure)
    print(f"7-src dequant 64M fp32 4-bit: {eff:.0f} GB/s")
    assert eff > 2500, f"multi-source dequant too slow: {eff:.0f} GB/s"
