"""In-tree build of the gfx950 HIP extension (_pa_hip.so).

Drives hipcc directly — no hipify, no CUDAExtension: the sources are native
HIP for gfx950 only. The .so lands inside the package directory so it
travels with repo snapshots (JIT caches under ~/.cache would not).

Usage:  python -m comfyui_parallelanything_amd.ops.build
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(PKG_DIR, "hip", "pa_ops.hip")
OUT = os.path.join(PKG_DIR, "_pa_hip.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build(verbose: bool = True) -> str:
    import torch
    from torch.utils import cpp_extension as ce

    torch_lib = ce.library_paths()[0]
    includes = ce.include_paths() + [sysconfig.get_paths()["include"]]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fno-gpu-rdc",
        SRC,
        "-o",
        OUT,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_pa_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
        "-DHIPBLAS_V2",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-Wno-deprecated-declarations",
        "-Wno-unused-result",
    ]
    cmd += [f"-I{p}" for p in includes]
    cmd += [
        f"-L{torch_lib}",
        f"-Wl,-rpath,{torch_lib}",
        "-ltorch",
        "-ltorch_hip",
        "-lc10",
        "-lc10_hip",
        "-ltorch_python",
        "-lamdhip64",
    ]
    if verbose:
        print("[build] " + " ".join(cmd))
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"[build] wrote {OUT}")
    return OUT


if __name__ == "__main__":
    build()
    # import check (host-side only; kernels need a GPU to run)
    sys.path.insert(0, os.path.dirname(os.path.dirname(PKG_DIR)))
    from comfyui_parallelanything_amd import ops

    ops._EXT_TRIED = False
    assert ops.hip_available(), "extension built but failed to load"
    print("[build] extension loads:", [
        n for n in dir(ops.hip_ext()) if not n.startswith("_")
    ])
