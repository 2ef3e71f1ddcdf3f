"""Within-process A/B timing of HIP kernel variants (run on a GPU box).

Times pbest_from_beta and eig_chunk at bench shapes for every
_coda_hip_v* module present, interleaving rounds (guide rule 24).
Also checks each variant's output against the base variant.
"""
import glob
import importlib
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

H, C, B = 128, 1000, 64
ROUNDS = 6


def main():
    dev = torch.device("cuda")
    mods = {}
    for so in sorted(glob.glob("coda_amd/ops/_coda_hip_v*.so")):
        name = os.path.basename(so).split(".")[0]
        mods[name.replace("_coda_hip_v", "")] = importlib.import_module(
            f"coda_amd.ops.{name}")
    print("variants:", list(mods))

    g = torch.Generator().manual_seed(0)
    a0 = (torch.rand(H, C, generator=g) * 50 + 1).to(dev)
    b0 = (torch.rand(H, C, generator=g) * 50 + 1).to(dev)
    cls = torch.randint(0, C, (B, H), generator=g).to(torch.int32).to(dev)
    pi = torch.softmax(torch.rand(C, generator=g), 0).to(dev)
    pixi = torch.softmax(torch.rand(B, C, generator=g), -1).to(dev)
    ref = None
    pb = None

    results = {k: [] for k in mods}
    outs = {}
    for k, m in mods.items():
        pb = m.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous(), 256)
        mix0 = (pi.unsqueeze(-1) * pb).sum(0)
        mm = mix0.clamp_min(1e-12)
        H0 = float(-(mm * mm.log2()).sum())
        outs[k] = (pb, mix0, H0)

    for rnd in range(ROUNDS):
        for k, m in mods.items():
            pb, mix0, H0 = outs[k]
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            eig = m.eig_chunk(a0, b0, cls, pb, pi, pixi, mix0, H0, 1.0, 256)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if rnd:
                results[k].append(dt)
            if ref is None:
                ref = eig
            elif rnd == 0:
                err = (eig - ref).abs().max()
                print(f"  {k}: max|eig - base| = {float(err):.3e}")

    rows = B * C
    for k, ts in results.items():
        med = sorted(ts)[len(ts) // 2]
        print(f"{k:10s} eig_chunk {1000*med:8.2f} ms  "
              f"({1e9*med/rows:6.1f} ns/row)  min {1000*min(ts):.2f}")


if __name__ == "__main__":
    main()
