"""CPU simulation of the compressed SRA allreduce: exactness on equal inputs
and the reference's analytic error bound (test_cgx.py:92 parity)."""

import numpy as np
import pytest
import torch

from torch_cgx_amd.parallel import sra_sim


@pytest.mark.parametrize("ws", [2, 4])
@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
@pytest.mark.parametrize("bits", [2, 4, 8])
def test_equal_inputs_exact(ws, dtype, bits):
    for n in [8, 128, 1024, 5000]:
        tensors = [torch.full((n,), float(3), dtype=dtype) for _ in range(ws)]
        out = sra_sim.sra_allreduce(tensors, [n], [(bits, 512)])
        expected = torch.full((n,), float(3 * ws), dtype=dtype)
        for o in out:
            assert torch.equal(o, expected), (n, bits)


@pytest.mark.parametrize("ws", [2, 4])
@pytest.mark.parametrize("bits", [2, 4, 8])
@pytest.mark.parametrize("bucket", [64, 512, 2048])
def test_error_bound_fp32(ws, bits, bucket):
    for n in [128, 1024, 1025, 16384]:
        arange = np.arange(-n / 2, n / 2, 1.0)
        tensors = [torch.tensor((r + 1) * arange, dtype=torch.float32)
                   for r in range(ws)]
        expected = torch.tensor((ws * (ws + 1) / 2) * arange,
                                dtype=torch.float32)
        out = sra_sim.sra_allreduce(tensors, [n], [(bits, bucket)])
        coef = ws * (ws + 1)
        bound = 2 * min(bucket, n) / ((1 << bits) - 1) * coef
        for o in out:
            err = (o - expected).abs().max().item()
            assert err < bound, (n, bits, bucket, err, bound)


def test_all_ranks_bitwise_identical():
    torch.manual_seed(0)
    ws, n = 4, 4096
    tensors = [torch.randn(n) for _ in range(ws)]
    out = sra_sim.sra_allreduce(tensors, [n], [(4, 512)])
    for o in out[1:]:
        assert torch.equal(o, out[0])


def test_multi_layer_mixed_config():
    torch.manual_seed(1)
    ws = 2
    layers = [1000, 513, 2048]
    cfgs = [(4, 512), (8, 64), (2, 1024)]
    n = sum(layers)
    tensors = [torch.randn(n) for _ in range(ws)]
    expected = sum(tensors)
    out = sra_sim.sra_allreduce(tensors, layers, cfgs)
    # loose bound: sum of per-layer bounds
    err = (out[0] - expected).abs().max().item()
    spread = max(t.abs().max().item() for t in tensors)
    bound = 2 * 1024 / 3 * ws * (ws + 1)  # dominated by the 2-bit layer
    assert err < bound
    assert torch.equal(out[0], out[1])


def test_tiny_chunks_more_ranks_than_elems():
    ws, n = 4, 5
    tensors = [torch.full((n,), float(r + 1)) for r in range(ws)]
    out = sra_sim.sra_allreduce(tensors, [n], [(4, 512)])
    expected = torch.full((n,), float(sum(range(1, ws + 1))))
    for o in out:
        assert torch.allclose(o, expected)


@pytest.mark.parametrize("ws", [3, 4])
def test_ring_equal_inputs_exact(ws):
    for n in [128, 4096]:
        tensors = [torch.full((n,), 2.5) for _ in range(ws)]
        out = sra_sim.ring_allreduce(tensors, [n], [(4, 512)])
        expected = torch.full((n,), 2.5 * ws)
        for o in out:
            assert torch.equal(o, expected)


@pytest.mark.parametrize("ws", [3, 4])
@pytest.mark.parametrize("bits", [4, 8])
def test_ring_error_bound(ws, bits):
    n, bucket = 16384, 512
    arange = np.arange(-n / 2, n / 2, 1.0)
    tensors = [torch.tensor((r + 1) * arange, dtype=torch.float32)
               for r in range(ws)]
    expected = torch.tensor((ws * (ws + 1) / 2) * arange, dtype=torch.float32)
    out = sra_sim.ring_allreduce(tensors, [n], [(bits, bucket)])
    # per-hop requantize: looser bound than SRA (ws-1 quantizations of
    # growing partial sums + the final one)
    bound = 2 * bucket / ((1 << bits) - 1) * ws * (ws + 1) * ws
    for o in out:
        err = (o - expected).abs().max().item()
        assert err < bound, (err, bound)
    for o in out[1:]:
        assert torch.equal(o, out[0])


def test_fuzz_random_configs():
    """Randomized layers/configs/world sizes: SRA result within an analytic
    bound of the exact sum, all ranks bitwise identical."""
    rng = np.random.default_rng(7)
    for trial in range(10):
        ws = int(rng.choice([2, 3, 4, 8]))
        nl = int(rng.integers(1, 5))
        layers = [int(rng.integers(1, 4000)) for _ in range(nl)]
        cfgs = [(int(rng.choice([2, 4, 8])), int(rng.choice([64, 512, 1024])))
                for _ in range(nl)]
        n = sum(layers)
        tensors = [torch.randn(n) for _ in range(ws)]
        out = sra_sim.sra_allreduce(tensors, layers, cfgs)
        exact = sum(tensors)
        # per-element bound: 2 * unit-scale * ws * (ws+1); use the max
        # range across ranks as the scale proxy
        scale = max(t.abs().max().item() for t in tensors) * 2
        worst_bits = min(b for b, _ in cfgs)
        bound = 2 * scale / ((1 << worst_bits) - 1) * ws * (ws + 1)
        err = (out[0] - exact).abs().max().item()
        assert err < bound, (trial, ws, layers, cfgs, err, bound)
        for o in out[1:]:
            assert torch.equal(o, out[0])


@pytest.mark.parametrize("ws", [2, 4])
def test_skip_incomplete_sim(ws):
    """Simulation with skip_incomplete: raw residual tails travel exactly."""
    torch.manual_seed(3)
    n, bits, bucket = 1000, 4, 512  # 1 full bucket + 488 raw residuals/chunk
    tensors = [torch.randn(n) for _ in range(ws)]
    out = sra_sim.sra_allreduce([t.clone() for t in tensors], [n],
                                [(bits, bucket)], skip=True)
    exact = sum(tensors)
    # all ranks identical; the residual region is reduced exactly
    for o in out[1:]:
        assert torch.equal(o, out[0])
    err = (out[0] - exact).abs()
    assert err.max().item() < 2 * bucket / 15 * ws * (ws + 1)


@pytest.mark.parametrize("nodes,local", [(2, 2), (2, 4), (4, 2)])
def test_hierarchical_compose_cpu(nodes, local):
    """The hierarchical composition (per-node SRA -> cross reduction over
    node results -> broadcast quantize round trip) stays within the stacked
    analytic bounds and is rank-identical — the CPU oracle the GPU
    hierarchical loopback test compares against, checked standalone."""
    from torch_cgx_amd.ops import golden
    torch.manual_seed(nodes * 10 + local)
    bits, bucket, n = 4, 512, 8192
    ws = nodes * local
    tensors = [torch.randn(n) for _ in range(ws)]
    exact = sum(tensors)

    node_res = []
    for nd in range(nodes):
        r = sra_sim.sra_allreduce(
            [tensors[nd * local + k].clone() for k in range(local)],
            [n], [(bits, bucket)])
        node_res.append(r[0])
    if nodes > 2:
        cross = sra_sim.ring_allreduce([t.clone() for t in node_res],
                                       [n], [(bits, bucket)])
    else:
        cross = sra_sim.sra_allreduce([t.clone() for t in node_res],
                                      [n], [(bits, bucket)])
    final = golden.dequantize(
        golden.quantize(cross[0], bits, bucket, rand=0.5), n,
        torch.float32, bits, bucket)

    # stacked bound: intra (local ranks) + cross (nodes, larger magnitudes)
    # + one broadcast quantization of the full sum
    unit_scale = 2 * bucket / ((1 << bits) - 1)
    bound = unit_scale * (local * (local + 1)
                          + local * nodes * (nodes + 1)
                          + ws + 1)
    err = (final - exact).abs().max().item()
    assert err < bound, (nodes, local, err, bound)
