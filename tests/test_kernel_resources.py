"""Kernel resource audit: compile the HIP source with
-Rpass-analysis=kernel-resource-usage and bound register/LDS/spill use.

Catches silent codegen regressions (occupancy cliffs, new scratch spills)
without a GPU - hipcc cross-compiles gfx950 anywhere. Loose bounds: the
tuned CODA_LB kernels intentionally spill ~50 VGPRs for 4 waves/SIMD
(measured faster - profiles/README.md); the glue kernels must not spill
at all.
"""
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HIPCC = "/opt/rocm/bin/hipcc"

pytestmark = pytest.mark.skipif(not os.path.exists(HIPCC),
                                reason="hipcc not available")


@pytest.fixture(scope="module")
def resource_report(tmp_path_factory):
    import torch.utils.cpp_extension as ce
    import sysconfig
    import torch
    out = tmp_path_factory.mktemp("kra") / "pbest.o"
    cmd = [HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-c", os.path.join(REPO, "coda_amd", "ops", "hip", "pbest.hip"),
           "-o", str(out), "-Rpass-analysis=kernel-resource-usage",
           "-DCODA_DPP_SCAN=1", "-DCODA_MIN_WAVES=4",
           "-DTORCH_EXTENSION_NAME=_kra", "-DTORCH_API_INCLUDE_EXTENSION_H",
           f"-D_GLIBCXX_USE_CXX11_ABI={int(torch.compiled_with_cxx11_abi())}",
           "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1"]
    cmd += [f"-I{p}" for p in ce.include_paths()]
    cmd += [f"-I{sysconfig.get_paths()['include']}"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    # parse blocks: "remark: <kernel> ... SGPRs: n ... VGPRs: n ...
    # ScratchSize [bytes/lane]: n ... Occupancy [waves/SIMD]: n"
    kernels = {}
    cur = None
    for line in r.stderr.splitlines():
        m = re.search(r"Function Name:\s*(\S+)", line)
        if m:
            cur = m.group(1)
            kernels[cur] = {}
        for key, pat in (("vgpr", r"VGPRs:\s*(\d+)"),
                         ("agpr", r"AGPRs:\s*(\d+)"),
                         ("scratch", r"ScratchSize \[bytes/lane\]:\s*(\d+)"),
                         ("occupancy", r"Occupancy \[waves/SIMD\]:\s*(\d+)"),
                         ("lds", r"LDS Size \[bytes/block\]:\s*(\d+)")):
            m = re.search(pat, line)
            if m and cur:
                kernels[cur][key] = int(m.group(1))
    assert kernels, "no resource remarks parsed:\n" + r.stderr[:2000]
    return kernels


def _find(kernels, substr):
    for name, info in kernels.items():
        if substr in name:
            return info
    raise AssertionError(f"kernel {substr} not found in {list(kernels)}")


def test_core_kernels_resource_envelope(resource_report):
    for k in ("pbest_kernel", "eig_hyp_kernel"):
        info = _find(resource_report, k)
        # tuned envelope: 4 waves/SIMD via <=128 VGPRs with bounded spill
        assert info["vgpr"] <= 128, (k, info)
        assert info.get("scratch", 0) <= 512, (k, info)
        assert info.get("occupancy", 4) >= 4, (k, info)


def test_glue_kernels_no_spill(resource_report):
    for k in ("es_build_kernel", "eig_assemble_kernel",
              "pi_hat_delta_kernel", "pi_marginal_kernel",
              "eig_totals_kernel", "eig_entropy_kernel"):
        info = _find(resource_report, k)
        assert info.get("scratch", 0) == 0, (k, info)
        assert info["vgpr"] <= 256, (k, info)
