"""Per-component timing of one CODA acquisition step at bench scale.

Run on a GPU box:  python scripts/perf_breakdown.py [--h 128 --n 50000 ...]
Prints wall time per component (device-synchronized).
"""
from __future__ import annotations

import argparse
import random
import sys
import time

import torch

sys.path.insert(0, ".")


def timed(fn, sync, reps=3):
    fn()  # warm
    sync()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    sync()
    return (time.perf_counter() - t0) / reps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--h", type=int, default=128)
    ap.add_argument("--n", type=int, default=50_000)
    ap.add_argument("--c", type=int, default=1000)
    ap.add_argument("--prefilter", type=int, default=256)
    ap.add_argument("--chunk", type=int, default=256)
    args = ap.parse_args()

    import os
    os.environ["CODA_BENCH_H"] = str(args.h)
    os.environ["CODA_BENCH_N"] = str(args.n)
    os.environ["CODA_BENCH_C"] = str(args.c)

    from coda_amd import CODA, Oracle
    from coda_amd.datasets import Dataset
    from coda_amd.options import LOSS_FNS
    from coda_amd import ops
    import bench

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    sync = (lambda: torch.cuda.synchronize()) if device.type == "cuda" \
        else (lambda: None)
    print(f"device={device} hip={ops.hip_available()}")

    preds, labels = bench.synth_preds(list(range(args.h)), args.n, args.c,
                                      device)
    ds = Dataset.from_tensors(preds, labels, device)
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(0); torch.manual_seed(0)

    t0 = time.perf_counter()
    sel = CODA(ds, prefilter_n=args.prefilter, chunk_size=args.chunk)
    sync()
    print(f"init                 {time.perf_counter()-t0:8.3f} s")

    print(f"update_pi_hat        {timed(sel.update_pi_hat, sync):8.3f} s")
    print(f"prefilter            "
          f"{timed(lambda: sel._prefilter(sel.unlabeled_idxs), sync):8.3f} s")
    print(f"pbest_rows_before    "
          f"{timed(sel._pbest_rows_before, sync):8.3f} s")

    pbest_before = sel._pbest_rows_before()
    alpha_cc, beta_cc = ops.dirichlet_to_beta(sel.dirichlets)
    mixture0, H_before = ops.mixture_entropy(pbest_before, sel.pi_hat)
    cand = torch.arange(args.chunk, device=device)
    chunk_classes = sel.classes[:, cand].t().contiguous()
    pi_xi = sel.pi_hat_xi[cand]

    if args.h <= 2048:  # v1 fused kernel path (eager fallback above that)
        t_chunk = timed(lambda: ops.eig_chunk(
            alpha_cc, beta_cc, chunk_classes, pbest_before, sel.pi_hat,
            pi_xi, mixture0, H_before), sync)
        rows = args.chunk * args.c
        print(f"eig_chunk B={args.chunk:5d}    {t_chunk:8.3f} s "
              f"({1e9*t_chunk/rows:.0f} ns/row, {rows/t_chunk/1e6:.2f} Mrow/s)")
    if device.type == "cuda":
        from coda_amd.ops import table as tops
        tables = tops.table_precompute(alpha_cc, beta_cc)
        t_tab = timed(lambda: tops.eig_chunk_table(
            tables, chunk_classes, pbest_before, sel.pi_hat, pi_xi,
            mixture0, H_before), sync)
        rows = args.chunk * args.c
        print(f"eig_chunk_table      {t_tab:8.3f} s "
              f"({1e9*t_tab/rows:.0f} ns/row)")

    print(f"eig_batched (full)   {timed(lambda: sel.eig_batched(), sync):8.3f} s")
    print(f"get_pbest            {timed(sel.get_pbest, sync):8.3f} s")

    def full_step():
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(int(idx)), q)
        sel.get_best_model_prediction()
    print(f"full step            {timed(full_step, sync, reps=2):8.3f} s")


if __name__ == "__main__":
    main()
