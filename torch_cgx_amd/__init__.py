"""torch_cgx_amd — MI355X-native compressed-allreduce backend for PyTorch.

A from-scratch re-design of the capabilities of IST-DASLab/torch_cgx for
AMD Instinct MI355X (gfx950): a ``torch.distributed`` backend named ``"cgx"``
whose fp32/fp16/bf16 SUM-allreduce of GPU tensors runs a 1-8 bit max-min
quantized Scatter-Reduce-AllGather over RCCL/xGMI, with hand-written CDNA4
HIP kernels for quantize / dequantize-accumulate.  Everything else
(broadcast, gather, scatter, alltoall, send/recv, barrier) is RCCL
passthrough on GPU and gloo on CPU.

Usage (drop-in for the reference torch_cgx):

    import torch_cgx_amd            # registers the "cgx" backend
    dist.init_process_group("cgx", init_method="env://", ...)
    model = DDP(model)
    state = torch_cgx_amd.CGXState(None, compression_params={"bits": 4})
    model.register_comm_hook(state, torch_cgx_amd.cgx_hook)

Launch with torchrun (one process per GPU) — no MPI anywhere.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

try:
    from . import _C  # noqa: F401  (built by setup.py build_ext --inplace)
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "torch_cgx_amd._C native extension is not built. "
        "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
        f"Underlying error: {e}") from e

__version__ = "0.1.0"


def _create_cgx(store, rank, size, timeout):
    """Backend creator registered with torch.distributed for name 'cgx'."""
    # CPU collectives delegate to an internal gloo backend (the reference used
    # MPI for these; gloo is the MPI-free equivalent on this stack).
    gloo = None
    try:
        from torch.distributed import ProcessGroupGloo
        from torch.distributed import PrefixStore
        gloo = ProcessGroupGloo(PrefixStore("cgx_gloo", store), rank, size, timeout)
    except Exception as e:  # gloo should always be available in torch-rocm
        import warnings
        warnings.warn(f"cgx: gloo CPU delegate unavailable: {e}")
    return _C.ProcessGroupCGX(store, rank, size, gloo)


def _register():
    if "cgx" in dist.Backend.backend_list:
        return
    dist.Backend.register_backend(
        "cgx", _create_cgx, devices=["cpu", "cuda"])


_register()

# ---- public config API (parity with the reference pybind surface,
#      /root/reference/src/ProcessGroupCGX.cc:852-857) ----
register_layer = _C.register_layer
set_quantization_bits = _C.set_quantization_bits
set_quantization_bucket_size = _C.set_quantization_bucket_size

from .hooks import CGXState, cgx_hook  # noqa: E402
from .adaptive import AdaptiveCGXState, adaptive_cgx_hook  # noqa: E402

__all__ = [
    "CGXState", "cgx_hook", "AdaptiveCGXState", "adaptive_cgx_hook",
    "register_layer", "set_quantization_bits",
    "set_quantization_bucket_size",
]
