"""Convert prediction-tensor files to a compact on-disk dtype.

The benchmark .pt files ship fp16/fp32; storing bf16 or fp8-e4m3 halves /
quarters both disk and host->device wire bytes (the loader up-casts to
fp32 for compute either way - coda_amd/datasets.py). Labels files are
copied unchanged.

Usage:
    python scripts/convert_task.py --task T [--data-dir data]
        [--dtype bf16|fp8] [--out-dir data_bf16]
    python scripts/convert_task.py --all --data-dir data --dtype fp8
"""
from __future__ import annotations

import argparse
import os
import shutil
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coda_amd.datasets import STORAGE_DTYPES  # noqa: E402


def convert(task: str, data_dir: str, out_dir: str, dtype: str):
    src = os.path.join(data_dir, task + ".pt")
    dst = os.path.join(out_dir, task + ".pt")
    t = torch.load(src, map_location="cpu", weights_only=True)
    t = t.to(STORAGE_DTYPES[dtype])
    os.makedirs(out_dir, exist_ok=True)
    torch.save(t, dst)
    lbl = src.replace(".pt", "_labels.pt")
    if os.path.exists(lbl):
        shutil.copyfile(lbl, dst.replace(".pt", "_labels.pt"))
    print(f"{task}: {os.path.getsize(src)/1e6:.1f} MB -> "
          f"{os.path.getsize(dst)/1e6:.1f} MB ({dtype})")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--task", default=None)
    ap.add_argument("--all", action="store_true")
    ap.add_argument("--data-dir", default="data")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--out-dir", default=None)
    args = ap.parse_args()
    out_dir = args.out_dir or f"{args.data_dir}_{args.dtype}"
    if args.all:
        tasks = sorted(f[:-3] for f in os.listdir(args.data_dir)
                       if f.endswith(".pt")
                       and not f.endswith("_labels.pt"))
    else:
        assert args.task, "--task or --all"
        tasks = [args.task]
    for t in tasks:
        convert(t, args.data_dir, out_dir, args.dtype)


if __name__ == "__main__":
    main()
