#!/usr/bin/env python3
"""In-tree build of the mi355x_ddp HIP extension for gfx950.

Drives hipcc DIRECTLY (no hipify, no CUDA shims — the sources are HIP-native)
and drops the .so inside the package (mi355x_ddp/_C.*.so) so the repo snapshot
carries it to GPU boxes. Cross-compiles fine on machines without a GPU.

Usage: python csrc/build.py [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOURCES = [os.path.join(REPO, "csrc", "mi355x_kernels.hip"),
           os.path.join(REPO, "csrc", "conv_igemm.hip")]


def out_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    return os.path.join(REPO, "mi355x_ddp", f"_C{suffix}")


def needs_build(out: str) -> bool:
    if not os.path.exists(out):
        return True
    out_mtime = os.path.getmtime(out)
    return any(os.path.getmtime(s) > out_mtime for s in SOURCES)


def build(force: bool = False, verbose: bool = True) -> str:
    import torch

    out = out_path()
    if not force and not needs_build(out):
        if verbose:
            print(f"[csrc/build] up to date: {out}")
        return out
    torch_dir = os.path.dirname(torch.__file__)
    inc = [
        f"{torch_dir}/include",
        f"{torch_dir}/include/torch/csrc/api/include",
        sysconfig.get_paths()["include"],
    ]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    cmd = (
        ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
         "-shared", "-DNDEBUG",
         f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
         "-DTORCH_EXTENSION_NAME=_C",
         "-DTORCH_API_INCLUDE_EXTENSION_H",
         "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
         "-DCUDA_HAS_FP16=1",
         "-D__HIP_NO_HALF_OPERATORS__=1", "-D__HIP_NO_HALF_CONVERSIONS__=1",
         "-fno-gpu-rdc", "-Wno-deprecated-declarations",
         "-Wno-unused-result"]
        + [f"-I{p}" for p in inc]
        + SOURCES
        + [f"-L{torch_dir}/lib", "-ltorch", "-ltorch_cpu", "-ltorch_hip",
           "-lc10", "-lc10_hip", "-ltorch_python",
           f"-Wl,-rpath,{torch_dir}/lib",
           "-o", out]
    )
    if verbose:
        print("[csrc/build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"[csrc/build] built {out}")
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
