"""Variant sweep for the pi_marginal kernel schedule (run on a GPU box).

Compiles scripts/pi_marginal_probe.hip on the box, checks every variant
against the torch fp32 reference, and times them at the headline shape
(N=50k, C=1000) plus the wide-pool shape. Prints one line per variant;
port only an evidenced winner into ops/hip/pbest.hip.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load
    here = os.path.dirname(os.path.abspath(__file__))
    ext = load(name="pim_probe", sources=[os.path.join(here, "pi_marginal_probe.hip")],
               extra_cflags=["-O3"], verbose=False)

    from coda_amd import ops
    assert ops.hip_available(), ops._ext_err
    dev = torch.device("cuda", 0)
    # vector-path probe variants require C % 4 == 0 (the shipping kernel
    # routes C % 4 != 0 to its scalar path, which is not the hot shape)
    for (N, C) in [(50_000, 1000), (1_000_000, 1000), (200_000, 128)]:
        torch.manual_seed(0)
        adjusted = torch.rand(N, C, device=dev) + 0.01
        row_sums = adjusted.sum(1)
        inv = 1.0 / row_sums.clamp_min(1e-12)
        ref = inv @ adjusted

        rows = []

        def check(name, out, us):
            err = (out - ref).abs().max().item() / ref.abs().max().item()
            rows.append((name, us, err))

        # shipping kernel (for a same-run baseline)
        check("ship(1536,r4)", ops._ext.pi_marginal(adjusted, row_sums),
              bench(lambda: ops._ext.pi_marginal(adjusted, row_sums)))
        for blocks in (256, 384, 512, 768):
            f = lambda b=blocks: ext.pim_atomic(adjusted, row_sums, b, 8)
            check(f"atomic({blocks},r8)", f(), bench(f))
        for G in (256, 512, 768):
            Cpad = (C + 3) & ~3
            partial = torch.empty(G, Cpad, device=dev)
            for gs in (1, 8, 16):
                f = lambda p=partial, s=gs: ext.pim_twostage(
                    adjusted, row_sums, p, 8, s)
                check(f"2stage({G},gs{gs})", f(), bench(f))
        # torch eager for reference
        check("torch-gemv", inv @ adjusted, bench(lambda: inv @ adjusted))

        print(f"== N={N} C={C} (traffic {N*C*4/1e6:.0f} MB, "
              f"floor ~{N*C*4/8e12*1e6:.0f} us) ==")
        for name, us, err in sorted(rows, key=lambda r: r[1]):
            print(f"  {name:18s} {us:8.1f} us   relerr={err:.2e}")
    sys.stdout.flush()


if __name__ == "__main__":
    main()
