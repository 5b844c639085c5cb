"""Step-scheduled quantization: an extension beyond the reference repo
(whose paper describes adaptive compression, but whose code ships only a
static per-layer registry).

``AdaptiveCGXState`` is a drop-in replacement for ``CGXState`` whose bit
width follows a step schedule — the standard recipe for low-bit gradient
training (full precision while gradients are large/ill-conditioned early on,
aggressive compression once training stabilizes):

    state = AdaptiveCGXState(None, schedule=[(0, 32), (50, 8), (500, 4)],
                             bucket_size=1024)
    model.register_comm_hook(state, cgx_hook)

At each scheduled boundary the native layer registry is re-written with the
new bit width (the layerwise filter still pins small/1-D layers at fp32).
"""

from typing import List, Optional, Sequence, Tuple

import torch.distributed as dist

from . import _C
from .hooks import CGXState, cgx_hook, VALUE_NO_COMPRESS  # noqa: F401


class AdaptiveCGXState(CGXState):
    def __init__(self, process_group: Optional[dist.ProcessGroup],
                 schedule: Sequence[Tuple[int, int]],
                 layer_min_size: int = 1024, bucket_size: int = 1024,
                 error_feedback: bool = False):
        params = {"bucket_size": bucket_size}
        if error_feedback:
            params["error_feedback"] = True
        sched = sorted(schedule)
        assert sched and sched[0][0] == 0, "schedule must start at step 0"
        for _, bits in sched:
            assert bits == VALUE_NO_COMPRESS or 1 <= bits <= 8
        params["bits"] = sched[0][1]
        super().__init__(process_group, layer_min_size=layer_min_size,
                         compression_params=params)
        self.schedule: List[Tuple[int, int]] = list(sched)
        self._applied_bits = sched[0][1]

    def _bits_for_step(self, step: int) -> int:
        bits = self.schedule[0][1]
        for s, b in self.schedule:
            if step >= s:
                bits = b
        return bits

    def on_bucket(self, bucket) -> None:
        """Apply the schedule: rewrite registered layer bit widths when a
        schedule boundary is crossed (fp32-pinned layers stay pinned)."""
        bits = self._bits_for_step(self.step)
        self.quantization_bits = bits  # used by step-2 registration
        if bits != self._applied_bits and self.step > 2:
            for (bucket_idx, _numels, cfgs) in _C.registry_snapshot():
                for layer_idx, (cur_bits, _bs) in enumerate(cfgs):
                    if cur_bits != VALUE_NO_COMPRESS:
                        _C.set_quantization_bits(bucket_idx, layer_idx, bits)
            self._applied_bits = bits


def adaptive_cgx_hook(state: AdaptiveCGXState, bucket):
    """cgx_hook with schedule application (same Future contract)."""
    state.on_bucket(bucket)
    return cgx_hook(state, bucket)


adaptive_cgx_hook.__annotations__ = dict(cgx_hook.__annotations__)
