"""DDP communication hook for the cgx backend.

API-compatible with the reference cgx_utils.allreduce_hooks
(/root/reference/cgx_utils/allreduce_hooks.py:29-73): ``CGXState`` carries the
process group and compression parameters; ``cgx_hook`` registers every
gradient tensor of every bucket with the native layer registry at step 2
(after DDP has finalized its buckets), then divides by world size and issues
an async SUM allreduce, returning its future.

Tensors with dim <= 1 (biases, LayerNorm/BatchNorm weights) or fewer than
``layer_min_size`` elements are registered with bits=32, i.e. excluded from
quantization — the "layerwise filter".
"""

import os
from typing import Dict, Optional

import torch
import torch.distributed as dist

from . import _C

COMPRESSION_QUANTIZATION_BITS = "CGX_COMPRESSION_QUANTIZATION_BITS"
COMPRESSION_BUCKET_SIZE = "CGX_COMPRESSION_BUCKET_SIZE"
COMPRESSION_MINIMAL_SIZE = "CGX_COMPRESSION_MINIMAL_SIZE"
VALUE_NO_COMPRESS = 32


class CGXState:
    def __init__(self, process_group: Optional[dist.ProcessGroup],
                 layer_min_size: int = 1024,
                 compression_params: Optional[Dict[str, int]] = None):
        self.process_group = (process_group if process_group is not None
                              else dist.group.WORLD)
        min_size_to_compress = int(os.getenv(COMPRESSION_MINIMAL_SIZE, "16"))
        self.layer_min_size = max(layer_min_size, min_size_to_compress)
        self.quantization_bits = int(
            os.getenv(COMPRESSION_QUANTIZATION_BITS, str(VALUE_NO_COMPRESS)))
        self.quantization_bucket_size = int(
            os.getenv(COMPRESSION_BUCKET_SIZE, "1024"))
        self.step = 0
        if compression_params is not None:
            self.quantization_bits = compression_params.get(
                "bits", self.quantization_bits)
            self.quantization_bucket_size = compression_params.get(
                "bucket_size", self.quantization_bucket_size)
            # engine config is env-driven (re-read every bucket), so every
            # rank picks this up consistently; write "0" explicitly when EF
            # is absent so a later CGXState doesn't inherit a stale "1"
            os.environ["CGX_ERROR_FEEDBACK"] = (
                "1" if compression_params.get("error_feedback") else "0")

    def should_compress_(self, tensor: torch.Tensor) -> bool:
        if tensor.dim() <= 1 or tensor.numel() < self.layer_min_size:
            return False
        return True


def _allreduce_fut(process_group: dist.ProcessGroup, tensor: torch.Tensor
                   ) -> torch.futures.Future[torch.Tensor]:
    group_to_use = (process_group if process_group is not None
                    else dist.group.WORLD)
    # Divide first to avoid fp16 overflow; the backend computes SUM.
    # (ws=1: division by 1 is the identity — skip the kernel launch)
    if group_to_use.size() > 1:
        tensor.div_(group_to_use.size())
    return (dist.all_reduce(tensor, group=group_to_use, async_op=True)
            .get_future()
            .then(lambda fut: fut.value()[0]))


def cgx_hook(state: CGXState, bucket) -> torch.futures.Future[torch.Tensor]:
    if state.step == 2:
        for layer_idx, tensor in enumerate(bucket.gradients()):
            bits = (state.quantization_bits if state.should_compress_(tensor)
                    else VALUE_NO_COMPRESS)
            _C.register_layer(bucket.index(), layer_idx, tensor.numel(),
                              bits, state.quantization_bucket_size)
    if bucket.is_last():
        state.step += 1
    return _allreduce_fut(state.process_group, bucket.buffer())
