"""Direct-hipcc build of torch_cgx_amd._C for gfx950.

No CUDAExtension / hipify machinery: the sources are hand-written HIP/C++,
compiled with hipcc against the torch-rocm headers and linked against torch,
RCCL and the HIP runtime.  Object files are cached by mtime under build/obj.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOURCES = [
    "csrc/quant_kernels.hip",
    "csrc/engine.cc",
    "csrc/transport.cc",
    "csrc/backend.cc",
    "csrc/bindings.cc",
]
HEADERS = ["csrc/compress.h", "csrc/engine.h", "csrc/backend.h",
           "csrc/transport.h"]
EXT_OUT = os.path.join(
    "torch_cgx_amd", "_C" + sysconfig.get_config_var("EXT_SUFFIX"))


def _mtime(p: str) -> float:
    return os.path.getmtime(p) if os.path.exists(p) else 0.0


def build(verbose: bool = True) -> str:
    import torch
    import torch.utils.cpp_extension as ce

    tinc = ce.include_paths()
    tlib = ce.library_paths()[0]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    flags = [
        "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H", "-DTORCH_EXTENSION_NAME=_C",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-Wno-deprecated-declarations",
    ]
    inc = [f"-I{p}" for p in tinc] + [
        f"-I{sysconfig.get_paths()['include']}",
        "-I/opt/rocm/include",
    ]
    objdir = os.path.join(ROOT, "build", "obj")
    os.makedirs(objdir, exist_ok=True)
    hdr_mtime = max(_mtime(os.path.join(ROOT, h)) for h in HEADERS)

    def compile_one(src: str) -> str:
        obj = os.path.join(objdir, os.path.basename(src) + ".o")
        src_abs = os.path.join(ROOT, src)
        if _mtime(obj) >= max(_mtime(src_abs), hdr_mtime):
            return obj
        extra = []
        if src.endswith(".hip"):
            # decode = round(unit*level) then round(min + product): FMA
            # contraction would fold this into one rounding and break
            # byte-parity with the golden wire-format model
            extra = ["-ffp-contract=off"]
        cmd = ["hipcc"] + flags + extra + inc + ["-c", src_abs, "-o", obj]
        if verbose:
            print("[cgx build]", src, flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            sys.stderr.write(r.stdout + r.stderr)
            raise RuntimeError(f"hipcc failed for {src}")
        return obj

    with ThreadPoolExecutor(max_workers=len(SOURCES)) as ex:
        objs = list(ex.map(compile_one, SOURCES))

    out = os.path.join(ROOT, EXT_OUT)
    if not os.path.exists(out) or any(_mtime(o) > _mtime(out) for o in objs):
        link = (["hipcc", "-shared", "-fPIC"] + objs + [
            f"-L{tlib}", "-ltorch", "-ltorch_cpu", "-ltorch_hip",
            "-ltorch_python", "-lc10", "-lc10_hip",
            "-L/opt/rocm/lib", "-lrccl", "-lamdhip64",
            f"-Wl,-rpath,{tlib}", "-Wl,-rpath,/opt/rocm/lib",
            "-o", out,
        ])
        if verbose:
            print("[cgx build] linking", EXT_OUT, flush=True)
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            sys.stderr.write(r.stdout + r.stderr)
            raise RuntimeError("link failed")
    return out


if __name__ == "__main__":
    build()
