"""Results table: cumulative regret x100 at step 100, per task x method.

Queries the RAW tracking-DB tables with the reference's exact SQL shape
(paper/tab1.py:28-51: metrics JOIN runs JOIN experiments JOIN tags on
'mlflow.parentRunId' / 'mlflow.runName'), averages over seeds, and prints
a table with the best and second-best method bolded per task
(paper/tab1.py:95-110).

Usage: python analysis/tab1.py [--db coda.sqlite] [--step 100]
               [--coda-name coda-lr=0.01-mult=2.0-no-prefilter] [--latex]
"""
from __future__ import annotations

import argparse
import sqlite3
from pathlib import Path

import pandas as pd

METHOD_LABELS = {
    "activetesting": "Active Testing",
    "iid": "Random Sampling",
    "model_picker": "Model Selector",
    "uncertainty": "Uncertainty",
    "vma": "VMA",
}


def extract_method_from_run_name(run_name: str) -> str:
    """Child runs are named '<task>-<method>[-<seed>]' (main.py run
    naming); recover the method string, which may itself contain dashes
    (e.g. 'coda-lr=0.01-mult=2.0-no-prefilter')."""
    segments = run_name.split("-")
    if segments[-1].isdigit() and len(segments) > 1:
        segments.pop()  # trailing seed number
    return "-".join(segments[1:]) or run_name


def load(db_path: str, metric: str, step: int) -> pd.DataFrame:
    db = Path(db_path).expanduser().resolve()
    if not db.exists():
        raise FileNotFoundError(f"Tracking DB not found: {db}")
    sql = f"""
    SELECT  e.name  AS task,
            rn.value AS run_name,
            m.value  AS value,
            m.step   AS step
    FROM    metrics m
    JOIN    runs r ON m.run_uuid = r.run_uuid
    JOIN    experiments e ON r.experiment_id = e.experiment_id
    JOIN    tags t_parent
           ON r.run_uuid = t_parent.run_uuid
          AND t_parent.key = 'mlflow.parentRunId'
    LEFT JOIN tags rn
           ON r.run_uuid = rn.run_uuid
          AND rn.key = 'mlflow.runName'
    WHERE   m.key = '{metric}'
      AND   m.step = {step}
      AND   r.lifecycle_stage = 'active'
      AND   e.lifecycle_stage = 'active'
    """
    with sqlite3.connect(str(db)) as conn:
        return pd.read_sql_query(sql, conn)


def build_table(db: str, metric: str = "cumulative regret",
                step: int = 100,
                coda_name: str = "coda-lr=0.01-mult=2.0-no-prefilter"):
    df = load(db, metric, step)
    if df.empty:
        return None, None
    df["method"] = df["run_name"].apply(extract_method_from_run_name)
    mean = df.groupby(["task", "method"], as_index=False)["value"].mean()
    std = df.groupby(["task", "method"], as_index=False)["value"].std()
    for frame in (mean, std):
        keep = (~frame.method.str.contains("coda")) | \
            (frame.method == coda_name)
        frame.drop(frame[~keep].index, inplace=True)
        for raw, label in METHOD_LABELS.items():
            frame.loc[frame.method == raw, "method"] = label
        frame.loc[frame.method == coda_name, "method"] = "CODA (Ours)"
        frame["value"] *= 100.0
    return (mean.pivot(index="task", columns="method", values="value"),
            std.pivot(index="task", columns="method", values="value"))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db", default="coda.sqlite")
    ap.add_argument("--metric", default="cumulative regret")
    ap.add_argument("--step", type=int, default=100)
    ap.add_argument("--coda-name",
                    default="coda-lr=0.01-mult=2.0-no-prefilter")
    ap.add_argument("--latex", action="store_true")
    args = ap.parse_args()

    table, _ = build_table(args.db, args.metric, args.step, args.coda_name)
    if table is None:
        print("No rows found (did you run main.py / aggregate yet?)")
        return
    if args.latex:
        lines = [" & ".join(["task"] + list(table.columns)) + r" \\"]
        for task, row in table.iterrows():
            vals = row.values
            order = vals.argsort()
            cells = []
            for i, v in enumerate(vals):
                s = f"{v:.2f}"
                if len(order) > 0 and i == order[0]:
                    s = r"\textbf{" + s + "}"
                elif len(order) > 1 and i == order[1]:
                    s = r"\underline{" + s + "}"
                cells.append(s)
            lines.append(" & ".join([str(task)] + cells) + r" \\")
        print("\n".join(lines))
    else:
        pd.set_option("display.width", 160)
        print(f"{args.metric} x100 at step {args.step} (mean over seeds; "
              "lower is better)")
        print(table.round(2).to_string())
        print("\nmean over tasks:")
        print(table.mean(axis=0).round(3).to_string())


if __name__ == "__main__":
    main()
