"""Local task-parallel launcher: tasks x methods across the GPUs of one node.

Replaces the reference's SLURM launcher (scripts/launch_all_methods.py:
one srun + 1 GPU per job, <=32 concurrent) with a single-node runner that
pins one harness process per GPU via HIP_VISIBLE_DEVICES - the
paper-reproduction config "all 26 tasks x 5 seeds across 8 GPUs"
(BASELINE.json config 4). No collectives: processes coordinate only
through the tracking DB, preserving the MLflow idempotence protocol
(run_needed <-> launch_all_methods.py:30-43).

Method strings may encode hyperparameters exactly as the reference's do
(`coda-lr=0.01-mult=2.0-no-prefilter`, parsed with the same patterns:
launch_all_methods.py:155-182), and downstream analysis parses the same
strings back out of run names.
"""
from __future__ import annotations

import argparse
import os
import re
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coda_amd import tracking  # noqa: E402


def seed_run_status(task, method, seed):
    runs = tracking.search_runs(
        experiment_names=[task],
        filter_string=f"tags.mlflow.runName = '{task}-{method}-{seed}'",
        max_results=1)
    if len(runs) == 0:
        return None, False
    finished = runs.status.values[0] == "FINISHED"
    stochastic = ("params.stochastic" in runs.columns and
                  runs["params.stochastic"].values[0] == "True")
    return finished, stochastic


def run_needed(task, method, max_seeds):
    finished0, stochastic0 = seed_run_status(task, method, 0)
    if finished0 is None:
        return True
    if not finished0:
        return True
    if not stochastic0:
        return False
    for seed in range(1, max_seeds):
        finished, _ = seed_run_status(task, method, seed)
        if not finished:
            return True
    return False


def method_to_args(method: str):
    """Decode hyperparameters from a method string (reference patterns)."""
    extra = []
    for pat, flag in [(r"-lr=([0-9.]+)", "--learning-rate"),
                      (r"-alpha=([0-9.]+)", "--alpha"),
                      (r"-mult=([0-9.]+)", "--multiplier"),
                      (r"-q=([a-z]+)", "--q")]:
        m = re.search(pat, method)
        if m:
            extra += [flag, m.group(1)]
    if "-no-prefilter" in method:
        extra += ["--prefilter-n", "0"]
    if "-no-diag" in method:
        extra += ["--no-diag-prior"]
    return extra


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--pred-dir", default="data")
    p.add_argument("--methods",
                   default="iid,activetesting,vma,model_picker,"
                           "uncertainty,coda-lr=0.01-mult=2.0-no-prefilter")
    p.add_argument("--seeds", type=int, default=5)
    p.add_argument("--gpus", type=int, default=None,
                   help="GPU slots to use (default: all visible, or 1 CPU "
                        "slot when no GPU)")
    p.add_argument("--max-concurrent", type=int, default=None)
    p.add_argument("--polling-interval", type=float, default=2.0)
    p.add_argument("--tasks", default="all")
    p.add_argument("--iters", type=int, default=100)
    p.add_argument("--dry-run", action="store_true")
    args = p.parse_args()

    import torch
    n_gpus = args.gpus
    if n_gpus is None:
        n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    slots = max(1, n_gpus)
    max_conc = args.max_concurrent or slots

    if args.tasks == "all":
        tasks = sorted(f[:-3] for f in os.listdir(args.pred_dir)
                       if f.endswith(".pt") and not f.endswith("_labels.pt"))
    else:
        tasks = args.tasks.split(",")
    methods = [m.strip() for m in args.methods.split(",") if m.strip()]

    queue = []
    for task in tasks:
        for method in methods:
            if not run_needed(task, method, args.seeds):
                print(f"Skipping {task}/{method}; all seeds finished")
                continue
            cmd = [sys.executable, "main.py", "--task", task,
                   "--method", method, "--data-dir", args.pred_dir,
                   "--seeds", str(args.seeds), "--iters", str(args.iters)]
            cmd += method_to_args(method)
            queue.append(cmd)

    if not queue:
        print("No jobs to run!")
        return
    print(f"{len(queue)} jobs over {slots} GPU slot(s), "
          f"max {max_conc} concurrent")
    if args.dry_run:
        for cmd in queue:
            print(" ".join(cmd))
        return

    running = {}  # slot -> (Popen, cmd)
    idx = 0
    free = list(range(max_conc))
    while idx < len(queue) or running:
        while idx < len(queue) and free:
            slot = free.pop(0)
            cmd = queue[idx]
            env = dict(os.environ)
            if n_gpus > 0:
                env["HIP_VISIBLE_DEVICES"] = str(slot % n_gpus)
                env["CUDA_VISIBLE_DEVICES"] = str(slot % n_gpus)
            print(f"[slot {slot}] launching: {' '.join(cmd[1:])}")
            proc = subprocess.Popen(cmd, env=env,
                                    stdout=subprocess.DEVNULL,
                                    stderr=subprocess.STDOUT)
            running[slot] = (proc, cmd)
            idx += 1
        time.sleep(args.polling_interval)
        for slot in list(running):
            proc, cmd = running[slot]
            rc = proc.poll()
            if rc is not None:
                status = "done" if rc == 0 else f"FAILED rc={rc}"
                print(f"[slot {slot}] {status}: {' '.join(cmd[3:5])}")
                del running[slot]
                free.append(slot)
        done = idx - len(running)
        print(f"Progress: {done}/{len(queue)} completed, "
              f"{len(running)} running, {len(queue) - idx} pending")
    print("All jobs completed!")


if __name__ == "__main__":
    main()
