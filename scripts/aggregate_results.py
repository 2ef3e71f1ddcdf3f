"""Aggregate child-run (seed) metrics onto parent (method) runs.

For every experiment and each parent run, compute the step-wise mean of
the chosen metrics across its child runs and log them back on the parent
as `mean_<metric>` (reference: scripts/aggregate_results.py:30-97).

Usage:
    python scripts/aggregate_results.py                 # regret metrics
    python scripts/aggregate_results.py m1 m2           # custom metrics
"""
from __future__ import annotations

import collections
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coda_amd import tracking  # noqa: E402


def aggregate_metrics(metric_keys=None):
    if metric_keys is None:
        metric_keys = ["regret", "cumulative regret"]

    for exp_id, exp_name in tracking.list_experiments():
        runs = tracking.search_runs(experiment_ids=[exp_id])
        if len(runs) == 0:
            continue
        has_parent_col = "tags.mlflow.parentRunId" in runs.columns
        for _, row in runs.iterrows():
            parent_tag = row.get("tags.mlflow.parentRunId") \
                if has_parent_col else None
            if isinstance(parent_tag, str):
                continue  # child run
            parent_id = row["run_id"]
            children = [r for _, r in runs.iterrows()
                        if has_parent_col and
                        r.get("tags.mlflow.parentRunId") == parent_id]
            if not children:
                continue
            for key in metric_keys:
                per_step = collections.defaultdict(list)
                for child in children:
                    for step, value in tracking.get_metric_history(
                            child["run_id"], key):
                        per_step[step].append(value)
                if not per_step:
                    continue
                for step in sorted(per_step):
                    vals = per_step[step]
                    tracking.log_metric_to_run(
                        parent_id, f"mean_{key}",
                        sum(vals) / len(vals), step=step)
                print(f"[{exp_name}] {row.get('tags.mlflow.runName', parent_id)}: "
                      f"mean_{key} over {len(children)} seeds, "
                      f"{len(per_step)} steps")


if __name__ == "__main__":
    keys = sys.argv[1:] or None
    aggregate_metrics(keys)
