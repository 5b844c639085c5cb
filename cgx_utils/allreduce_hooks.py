"""Drop-in replacement for the reference cgx_utils.allreduce_hooks."""
from torch_cgx_amd.hooks import (  # noqa: F401
    CGXState, cgx_hook, _allreduce_fut,
    COMPRESSION_QUANTIZATION_BITS, COMPRESSION_BUCKET_SIZE,
    COMPRESSION_MINIMAL_SIZE, VALUE_NO_COMPRESS,
)
