"""Per-step latency report from the tracking DB's step_seconds metric.

The harness logs wall time of every acquisition step (new telemetry, no
reference counterpart - SURVEY.md section 5.1 gap). This prints per
(task, method) latency percentiles and the implied steps/sec.

Usage: python analysis/latency.py [--db coda.sqlite]
"""
from __future__ import annotations

import argparse
import sqlite3
from pathlib import Path

import numpy as np
import pandas as pd

from tab1 import extract_method_from_run_name


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--db", default="coda.sqlite")
    args = ap.parse_args()

    sql = """
    SELECT e.name AS task, rn.value AS run_name, m.value AS secs
    FROM metrics m
    JOIN runs r ON m.run_uuid = r.run_uuid
    JOIN experiments e ON r.experiment_id = e.experiment_id
    LEFT JOIN tags rn ON r.run_uuid = rn.run_uuid
         AND rn.key = 'mlflow.runName'
    WHERE m.key = 'step_seconds'
      AND r.lifecycle_stage = 'active' AND e.lifecycle_stage = 'active'
    """
    with sqlite3.connect(str(Path(args.db).resolve())) as conn:
        df = pd.read_sql_query(sql, conn)
    if df.empty:
        print("no step_seconds metrics (runs logged with --no-mlflow?)")
        return
    df["method"] = df["run_name"].apply(extract_method_from_run_name)
    print(f"{'task':22s} {'method':36s} {'p50 ms':>8s} {'p90 ms':>8s} "
          f"{'max ms':>8s} {'steps/s':>8s} {'n':>5s}")
    for (task, method), g in df.groupby(["task", "method"]):
        v = g.secs.values * 1000
        print(f"{task:22s} {method:36s} {np.percentile(v, 50):8.2f} "
              f"{np.percentile(v, 90):8.2f} {v.max():8.2f} "
              f"{1000.0 / np.percentile(v, 50):8.1f} {len(v):5d}")


if __name__ == "__main__":
    main()
