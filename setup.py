#!/usr/bin/env python3
"""Build torch_cgx_amd's native extension in-tree.

Usage: python setup.py build_ext --inplace   (any argument form builds)
The build is a direct hipcc invocation (tools/build_native.py) — the sources
are hand-written HIP/CDNA4, so no CUDAExtension/hipify pass is involved.
"""

import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "tools"))
import build_native  # noqa: E402

if __name__ == "__main__":
    build_native.build()
    print("torch_cgx_amd._C built in-tree")
