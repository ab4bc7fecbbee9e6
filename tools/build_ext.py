"""In-tree build of the gfx950 HIP extension.

Drives hipcc DIRECTLY (no hipify, no CUDA-compat pass): csrc/ops.hip is
native HIP/CDNA4 code.  The resulting spark_ensemble_amd/_hip_ops.so lives
in-tree so it travels to GPU boxes with the repo snapshot (it is
git-ignored; gpurun ships it).

Usage:  python tools/build_ext.py   (or via setup.py / __graft_entry__.build)
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = [
    os.path.join(REPO, "csrc", "ops.hip"),
    os.path.join(REPO, "csrc", "linear.hip"),
]
OUT = os.path.join(REPO, "spark_ensemble_amd", "_hip_ops.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build(verbose: bool = True) -> str:
    import torch
    import torch.utils.cpp_extension as ce

    includes = ce.include_paths(device_type="cuda") + [
        sysconfig.get_paths()["include"]
    ]
    libs = ce.library_paths(device_type="cuda")
    abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"

    newest_src = max(os.path.getmtime(s) for s in SRC)
    if os.path.exists(OUT) and os.path.getmtime(OUT) > newest_src:
        if verbose:
            print(f"[build_ext] up to date: {OUT}")
        return OUT

    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-munsafe-fp-atomics",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DCUDA_HAS_FP16=1",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fno-gpu-rdc",
    ]
    for i in includes:
        cmd += ["-I", i]
    cmd += SRC
    for l in libs:
        cmd += ["-L", l, f"-Wl,-rpath,{l}"]
    cmd += [
        "-ltorch",
        "-ltorch_cpu",
        "-ltorch_hip",
        "-lc10",
        "-lc10_hip",
        "-ltorch_python",
        "-lamdhip64",
        "-o",
        OUT,
    ]
    if verbose:
        print("[build_ext]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build()
