"""Headline figure: fraction of benchmarks converged to <1% regret vs
number of labels (reference: paper/fig1.py:60-118).

For each (task, method): seed-mean regret per step; convergence step = the
first step s such that regret stays below THRESHOLD for every later step;
then the per-step fraction of tasks converged. Writes fig1.png and prints
the per-method convergence fractions at steps 10/25/50/100.

Usage: python analysis/fig1.py [--db coda.sqlite] [--out fig1.png]
"""
from __future__ import annotations

import argparse
import sqlite3
from pathlib import Path

import numpy as np
import pandas as pd

from tab1 import METHOD_LABELS, extract_method_from_run_name

THRESHOLD = 1.0      # percent regret
MAX_STEPS = 100
NO_CONVERGENCE = 999


def load_regrets(db_path: str) -> pd.DataFrame:
    db = Path(db_path).expanduser().resolve()
    sql = """
    SELECT  e.name AS task, rn.value AS run_name, m.value AS regret,
            m.step AS step
    FROM    metrics m
    JOIN    runs r ON m.run_uuid = r.run_uuid
    JOIN    experiments e ON r.experiment_id = e.experiment_id
    JOIN    tags t_parent ON r.run_uuid = t_parent.run_uuid
           AND t_parent.key = 'mlflow.parentRunId'
    LEFT JOIN tags rn ON r.run_uuid = rn.run_uuid
           AND rn.key = 'mlflow.runName'
    WHERE   m.key = 'regret'
      AND   r.lifecycle_stage = 'active' AND e.lifecycle_stage = 'active'
    """
    with sqlite3.connect(str(db)) as conn:
        return pd.read_sql_query(sql, conn)


def convergence_fractions(db: str,
                          coda_name="coda-lr=0.01-mult=2.0-no-prefilter"):
    df = load_regrets(db)
    if df.empty:
        return None, None
    df["method"] = df["run_name"].apply(extract_method_from_run_name)
    keep = (~df.method.str.contains("coda")) | (df.method == coda_name)
    df = df[keep].copy()
    for raw, label in METHOD_LABELS.items():
        df.loc[df.method == raw, "method"] = label
    df.loc[df.method == coda_name, "method"] = "CODA (Ours)"
    df["regret"] *= 100.0

    mean = df.groupby(["task", "method", "step"],
                      as_index=False)["regret"].mean()
    tasks = mean.task.unique()
    methods = mean.method.unique()

    conv = {m: {} for m in methods}
    for method in methods:
        for task in tasks:
            series = mean[(mean.task == task) & (mean.method == method)] \
                .sort_values("step").regret.tolist()
            step = NO_CONVERGENCE
            for start in range(min(MAX_STEPS, len(series))):
                if all(v < THRESHOLD for v in series[start:]):
                    step = start + 1
                    break
            conv[method][task] = step

    fractions = {}
    for method in methods:
        frac = np.zeros(MAX_STEPS)
        for s in range(1, MAX_STEPS + 1):
            n = sum(1 for t in tasks
                    if conv[method][t] != NO_CONVERGENCE
                    and conv[method][t] <= s)
            frac[s - 1] = n / len(tasks)
        fractions[method] = frac
    return fractions, tasks


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db", default="coda.sqlite")
    ap.add_argument("--out", default="fig1.png")
    args = ap.parse_args()

    fractions, tasks = convergence_fractions(args.db)
    if fractions is None:
        print("No regret metrics found.")
        return
    print(f"{len(tasks)} tasks; fraction converged to <{THRESHOLD}% regret:")
    print(f"{'method':18s} " + " ".join(f"@{s:<4d}" for s in (10, 25, 50, 100)))
    for method, frac in sorted(fractions.items()):
        print(f"{method:18s} " +
              " ".join(f"{frac[s-1]:5.2f}" for s in (10, 25, 50, 100)))

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    fig, ax = plt.subplots(figsize=(5.5, 5))
    for method, frac in sorted(fractions.items()):
        ax.plot(range(1, MAX_STEPS + 1), frac, label=method)
    ax.set_xlabel("number of labels")
    ax.set_ylabel(f"fraction of benchmarks < {THRESHOLD}% regret")
    ax.legend(fontsize=8)
    ax.set_ylim(0, 1)
    plt.tight_layout()
    plt.savefig(args.out, dpi=150)
    print("wrote", args.out)


if __name__ == "__main__":
    main()
