"""Build the gfx950 HIP extension in-tree.

Invokes hipcc directly (no hipify, no CUDA shims): the kernel sources in
coda_amd/ops/hip/ are native HIP/CDNA4 code. The resulting
coda_amd/ops/_coda_hip.so is imported by coda_amd.ops and travels with the
repo snapshot to GPU machines (it is git-ignored but not gpurun-ignored).

Usage: python build_hip.py [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))
SRC = [os.path.join(REPO, "coda_amd", "ops", "hip", "pbest.hip"),
       os.path.join(REPO, "coda_amd", "ops", "hip", "pair.hip")]
OUT = os.path.join(REPO, "coda_amd", "ops",
                   "_coda_hip.cpython-310-x86_64-linux-gnu.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build(force: bool = False, verbose: bool = True) -> str:
    import torch
    import torch.utils.cpp_extension as ce

    if not force and os.path.exists(OUT):
        newest_src = max(os.path.getmtime(s) for s in SRC + [__file__])
        if os.path.getmtime(OUT) >= newest_src:
            if verbose:
                print(f"[build_hip] build_mode=reuse (up to date): {OUT}")
            return OUT
    if verbose:
        print("[build_hip] build_mode=compile")

    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    includes = ce.include_paths() + [sysconfig.get_paths()["include"]]
    libdirs = ce.library_paths()
    abi = int(torch.compiled_with_cxx11_abi())

    # Tuned on MI355X (scripts/kernel_bench.py sweep): DPP wave scan +
    # 4-waves/SIMD launch bound = 1.86x the untuned kernel (82.9 -> 44.6
    # ns/row on the fused EIG kernel at H=128, C=1000).
    cmd = [hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
           "-fPIC", "-shared", *SRC, "-o", OUT,
           "-DCODA_DPP_SCAN=1", "-DCODA_MIN_WAVES=4",
           "-DTORCH_EXTENSION_NAME=_coda_hip",
           "-DTORCH_API_INCLUDE_EXTENSION_H",
           f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
           "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
           "-fno-gpu-rdc", "-Wno-deprecated-declarations"]
    cmd += [f"-I{p}" for p in includes]
    cmd += [f"-L{p}" for p in libdirs]
    cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
            "-ltorch_python", "-lamdhip64"]
    if verbose:
        print("[build_hip]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"[build_hip] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
