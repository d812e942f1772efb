#!/usr/bin/env python3
"""In-tree build of the native extension: hipcc --offload-arch=gfx950 over
native/ext.hip -> dist_mnist_amd/_C.so. No JIT cache: the .so lives in the
repo tree so it travels with gpurun snapshots.

Flags are derived from torch.utils.cpp_extension so the ABI matches the
installed PyTorch-ROCm exactly, but the compile itself is a single explicit
hipcc invocation (no hipify pass — the sources are native HIP already).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def build(verbose: bool = True, asan: bool = False) -> str:
    import torch
    import torch.utils.cpp_extension as ce

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    rocm = os.environ.get("ROCM_HOME", "/opt/rocm")
    hipcc = os.path.join(rocm, "bin", "hipcc")
    out = os.path.join(REPO, "dist_mnist_amd", "_C.so")
    src = os.path.join(REPO, "native", "ext.hip")

    incs = ce.include_paths(device_type="cuda")
    libdirs = ce.library_paths(device_type="cuda")
    py_inc = sysconfig.get_paths()["include"]

    cmd = [
        hipcc, "-O3", "-std=c++17", "--offload-arch=gfx950", "-fPIC", "-shared",
        src, "-o", out,
        f"-I{py_inc}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={int(torch._C._GLIBCXX_USE_CXX11_ABI)}",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    if asan:
        # Host-side AddressSanitizer (SURVEY.md §5 race/sanitizer plan):
        # instruments the C++ binding/host code only (-Xarch_host keeps the
        # gfx950 device pass clean). Run python under the matching runtime:
        #   LD_PRELOAD=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so) \
        #   ASAN_OPTIONS=detect_leaks=0 python -m pytest ...
        # (leak detection off: torch/HIP runtime allocations dwarf ours).
        cmd += ["-Xarch_host", "-fsanitize=address", "-shared-libsan", "-g"]
    cmd += list(getattr(ce, "COMMON_HIP_FLAGS", []))
    cmd += [f"-I{i}" for i in incs]
    cmd += [f"-L{d}" for d in libdirs]
    cmd += ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python",
            "-lamdhip64", "-lrccl"]
    if verbose:
        print(" ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(asan="--asan" in sys.argv)
    print("built dist_mnist_amd/_C.so")
    sys.exit(0)
