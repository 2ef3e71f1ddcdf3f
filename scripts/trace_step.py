"""Tracing utility: capture one acquisition step as a Chrome trace.

The framework's kernels are named (rocprofv3-visible); this adds the
host-side view: a torch.profiler capture of N steps at a chosen config,
written as chrome://tracing JSON plus a top-op table.

Usage (GPU box):
    python scripts/trace_step.py [--h 128 --n 50000 --c 1000]
        [--steps 3] [--out gpurun_out/trace.json]
"""
from __future__ import annotations

import argparse
import random
import sys

sys.path.insert(0, ".")

import torch
from torch.profiler import ProfilerActivity, profile


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--h", type=int, default=128)
    ap.add_argument("--n", type=int, default=50_000)
    ap.add_argument("--c", type=int, default=1000)
    ap.add_argument("--prefilter", type=int, default=256)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--out", default="gpurun_out/trace.json")
    args = ap.parse_args()

    from coda_amd import CODA, Oracle
    from coda_amd.datasets import Dataset
    from coda_amd.options import LOSS_FNS
    import bench

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    preds, labels = bench.synth_preds(list(range(args.h)), args.n, args.c,
                                      device)
    ds = Dataset.from_tensors(preds, labels, device)
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, prefilter_n=args.prefilter, chunk_size=args.prefilter)

    def step():
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(int(idx)), q)
        sel.get_best_model_prediction()

    step()  # warm (tables built)
    if device.type == "cuda":
        torch.cuda.synchronize()
    acts = [ProfilerActivity.CPU]
    if device.type == "cuda":
        acts.append(ProfilerActivity.CUDA)
    with profile(activities=acts, record_shapes=True) as prof:
        for _ in range(args.steps):
            step()
        if device.type == "cuda":
            torch.cuda.synchronize()
    prof.export_chrome_trace(args.out)
    sort = "cuda_time_total" if device.type == "cuda" else "cpu_time_total"
    print(prof.key_averages().table(sort_by=sort, row_limit=25))
    print("wrote", args.out)


if __name__ == "__main__":
    main()
