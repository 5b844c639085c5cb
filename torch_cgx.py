"""Import-name compatibility shim: `import torch_cgx` works like the reference.

Importing this module registers the "cgx" backend and re-exports the
reference's pybind surface (register_layer, set_quantization_bits,
set_quantization_bucket_size) from torch_cgx_amd.
"""
from torch_cgx_amd import (  # noqa: F401
    register_layer, set_quantization_bits, set_quantization_bucket_size,
    CGXState, cgx_hook,
)
