"""Failure-case analysis: where does a method pick the wrong model?

(Reference: paper/fig4.py - confusion matrices of failure cases.) For each
(task, method): the final regret per seed, the fraction of seeds ending on
a non-best model, and - given the task's .pt data - the per-class accuracy
difference between the truly-best model and the typical wrongly-picked
one (which classes fooled the selector).

Usage: python analysis/fig4.py [--db coda.sqlite] [--data-dir data]
"""
from __future__ import annotations

import argparse
import os
import sqlite3
from pathlib import Path

import pandas as pd
import torch

from tab1 import METHOD_LABELS, extract_method_from_run_name


def final_regrets(db_path: str) -> pd.DataFrame:
    sql = """
    SELECT e.name AS task, rn.value AS run_name, m.value AS regret,
           MAX(m.step) AS step
    FROM metrics m
    JOIN runs r ON m.run_uuid = r.run_uuid
    JOIN experiments e ON r.experiment_id = e.experiment_id
    JOIN tags t_parent ON r.run_uuid = t_parent.run_uuid
         AND t_parent.key = 'mlflow.parentRunId'
    LEFT JOIN tags rn ON r.run_uuid = rn.run_uuid
         AND rn.key = 'mlflow.runName'
    WHERE m.key = 'regret'
      AND r.lifecycle_stage = 'active' AND e.lifecycle_stage = 'active'
    GROUP BY m.run_uuid
    """
    with sqlite3.connect(str(Path(db_path).resolve())) as conn:
        return pd.read_sql_query(sql, conn)


def per_class_gap(task_path: str):
    """(best_model, per-class accuracy of best vs pool median)."""
    preds = torch.load(task_path, map_location="cpu",
                       weights_only=True).float()
    labels_p = task_path.replace(".pt", "_labels.pt")
    if not os.path.exists(labels_p):
        return None
    labels = torch.load(labels_p, map_location="cpu", weights_only=True)
    H, N, C = preds.shape
    cls = preds.argmax(-1)
    acc = (cls == labels).float().mean(1)
    best = int(acc.argmax())
    per_class = torch.zeros(H, C)
    for c in range(C):
        m = labels == c
        if m.any():
            per_class[:, c] = (cls[:, m] == c).float().mean(1)
    return best, acc, per_class


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db", default="coda.sqlite")
    ap.add_argument("--data-dir", default="data")
    ap.add_argument("--coda-name",
                    default="coda-lr=0.01-mult=2.0-no-prefilter")
    args = ap.parse_args()

    df = final_regrets(args.db)
    if df.empty:
        print("No regret metrics found.")
        return
    df["method"] = df["run_name"].apply(extract_method_from_run_name)
    keep = (~df.method.str.contains("coda")) | (df.method == args.coda_name)
    df = df[keep].copy()
    for raw, label in METHOD_LABELS.items():
        df.loc[df.method == raw, "method"] = label
    df.loc[df.method == args.coda_name, "method"] = "CODA (Ours)"

    df["failed"] = df.regret > 1e-9
    summary = df.groupby(["task", "method"]).agg(
        fail_rate=("failed", "mean"),
        mean_final_regret=("regret", "mean")).reset_index()
    pd.set_option("display.width", 140)
    print("Final-step failure rate (fraction of seeds not on the best "
          "model) and mean final regret:")
    print(summary.pivot(index="task", columns="method",
                        values="fail_rate").round(2).to_string())

    # per-class anatomy of the hardest task for each failing method
    for task in sorted(df.task.unique()):
        p = os.path.join(args.data_dir, task + ".pt")
        if not os.path.exists(p):
            continue
        info = per_class_gap(p)
        if info is None:
            continue
        best, acc, per_class = info
        runner = int(acc.argsort(descending=True)[1])
        gap = per_class[best] - per_class[runner]
        worst_c = int(gap.argmin())
        print(f"\n{task}: best model {best} (acc {acc[best]:.3f}) vs "
              f"runner-up {runner} (acc {acc[runner]:.3f}); "
              f"runner-up beats best on class {worst_c} by "
              f"{-gap[worst_c]:.3f} - the class most likely to mislead "
              "selection")


if __name__ == "__main__":
    main()
