"""Per-task memory table + per-class-count group medians.

(Reference: paper/fig3.py - groups tasks by class count, reports the
prediction-tensor memory per task.) Reads the actual .pt files instead of
a hard-coded dict.

Usage: python analysis/fig3.py [--data-dir data] [--db coda.sqlite]
"""
from __future__ import annotations

import argparse
import os

import torch


def memory_table(data_dir: str):
    rows = []
    for f in sorted(os.listdir(data_dir)):
        if not f.endswith(".pt") or f.endswith("_labels.pt"):
            continue
        path = os.path.join(data_dir, f)
        t = torch.load(path, map_location="meta", weights_only=True,
                       mmap=False)
        H, N, C = t.shape
        gb = H * N * C * 4 / 1e9  # fp32 compute footprint
        rows.append((f[:-3], H, N, C, gb,
                     os.path.getsize(path) / 1e9))
    return rows


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--data-dir", default="data")
    args = ap.parse_args()

    rows = memory_table(args.data_dir)
    if not rows:
        print("no .pt tasks under", args.data_dir)
        return
    print(f"{'task':24s} {'H':>5s} {'N':>8s} {'C':>5s} "
          f"{'fp32 GB':>8s} {'disk GB':>8s}")
    for task, H, N, C, gb, disk in rows:
        print(f"{task:24s} {H:5d} {N:8d} {C:5d} {gb:8.3f} {disk:8.3f}")

    # group by class count (the reference's fig3 grouping)
    groups = {}
    for task, H, N, C, gb, _ in rows:
        groups.setdefault(C, []).append(gb)
    print("\nby class count:")
    for C in sorted(groups):
        vals = sorted(groups[C])
        med = vals[len(vals) // 2]
        print(f"  C={C:5d}: {len(vals)} tasks, median fp32 {med:.3f} GB")


if __name__ == "__main__":
    main()
