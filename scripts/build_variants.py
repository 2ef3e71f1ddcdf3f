"""Build tuning variants of the HIP extension as separate modules.

Each variant is the same pbest.hip compiled with different -D flags into
coda_amd/ops/_coda_hip_v<name>.so; scripts/kernel_bench.py times whichever
variants exist. Used for within-sweep A/B kernel tuning on a GPU box.
"""
import os
import subprocess
import sys
import sysconfig

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

VARIANTS = {
    "base": [],
    "lb4": ["-DCODA_MIN_WAVES=4"],
    "lb3": ["-DCODA_MIN_WAVES=3"],
    "unroll2": ["-DCODA_UNROLL_H=2"],
    "dpp": ["-DCODA_DPP_SCAN=1"],
    "dpp4": ["-DCODA_DPP_SCAN=1", "-DCODA_MIN_WAVES=4"],
    "dpp4u2": ["-DCODA_DPP_SCAN=1", "-DCODA_MIN_WAVES=4",
               "-DCODA_UNROLL_H=2"],
}


def build_all():
    import torch
    import torch.utils.cpp_extension as ce
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    src = os.path.join(repo, "coda_amd", "ops", "hip", "pbest.hip")
    includes = ce.include_paths() + [sysconfig.get_paths()["include"]]
    libdirs = ce.library_paths()
    abi = int(torch.compiled_with_cxx11_abi())
    for name, flags in VARIANTS.items():
        mod = f"_coda_hip_v{name}"
        out = os.path.join(repo, "coda_amd", "ops",
                           f"{mod}.cpython-310-x86_64-linux-gnu.so")
        cmd = ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-O3",
               "-std=c++17", "-fPIC", "-shared", src, "-o", out,
               f"-DTORCH_EXTENSION_NAME={mod}",
               "-DTORCH_API_INCLUDE_EXTENSION_H",
               f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
               "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
               "-Wno-deprecated-declarations", *flags]
        cmd += [f"-I{p}" for p in includes] + [f"-L{p}" for p in libdirs]
        cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10",
                "-lc10_hip", "-ltorch_python", "-lamdhip64"]
        print("building", mod, flags)
        subprocess.run(cmd, check=True)


if __name__ == "__main__":
    build_all()
