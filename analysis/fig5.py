"""Per-task regret curves: seed-mean regret vs step for every task/method.

(Reference: paper/fig5.py - a grid of per-task curves.) Writes one PNG
with a subplot per task.

Usage: python analysis/fig5.py [--db coda.sqlite] [--out fig5.png]
"""
from __future__ import annotations

import argparse
import math

from fig1 import load_regrets
from tab1 import METHOD_LABELS, extract_method_from_run_name


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db", default="coda.sqlite")
    ap.add_argument("--out", default="fig5.png")
    ap.add_argument("--coda-name",
                    default="coda-lr=0.01-mult=2.0-no-prefilter")
    args = ap.parse_args()

    df = load_regrets(args.db)
    if df.empty:
        print("No regret metrics found.")
        return
    df["method"] = df["run_name"].apply(extract_method_from_run_name)
    keep = (~df.method.str.contains("coda")) | (df.method == args.coda_name)
    df = df[keep].copy()
    for raw, label in METHOD_LABELS.items():
        df.loc[df.method == raw, "method"] = label
    df.loc[df.method == args.coda_name, "method"] = "CODA (Ours)"
    mean = df.groupby(["task", "method", "step"],
                      as_index=False)["regret"].mean()

    tasks = sorted(mean.task.unique())
    cols = min(4, len(tasks))
    rows = math.ceil(len(tasks) / cols)

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    fig, axes = plt.subplots(rows, cols, figsize=(3.2 * cols, 2.6 * rows),
                             squeeze=False)
    for i, task in enumerate(tasks):
        ax = axes[i // cols][i % cols]
        sub = mean[mean.task == task]
        for method in sorted(sub.method.unique()):
            s = sub[sub.method == method].sort_values("step")
            ax.plot(s.step, 100 * s.regret, label=method, linewidth=1)
        ax.set_title(task, fontsize=9)
        ax.set_ylim(bottom=0)
    axes[0][0].legend(fontsize=6)
    for ax in axes[-1]:
        ax.set_xlabel("labels")
    for row in axes:
        row[0].set_ylabel("regret (%)")
    plt.tight_layout()
    plt.savefig(args.out, dpi=150)
    print("wrote", args.out)


if __name__ == "__main__":
    main()
