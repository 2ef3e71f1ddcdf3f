"""Unsupervised epsilon tuning for the ModelPicker baseline.

Reference protocol (scripts/modelselector/modelselector_eps_gridsearch_v2.py):
for each candidate epsilon, run many random realisations on subsampled
pools using MAJORITY-VOTE pseudo-labels as a label-free oracle; score each
epsilon by (a) the success rate of identifying a pseudo-best model within
budget and (b) the earliest step at which it stays identified; write the
winners to best_epsilons.json.

Usage:
    python scripts/modelpicker_eps_gridsearch.py --task T --data-dir data
        [--eps 0.35:0.49:0.01] [--realisations 100] [--pool 1000]
        [--budget 100] [--threshold 0.9] [--out best_epsilons.json]
"""
from __future__ import annotations

import argparse
import json
import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coda_amd.datasets import Dataset  # noqa: E402
from coda_amd.baselines import ModelPicker  # noqa: E402
from coda_amd import ops  # noqa: E402


def majority_vote_labels(preds: torch.Tensor) -> torch.Tensor:
    """(H, N, C) -> (N,) majority vote of per-model argmaxes (the
    label-free pseudo-oracle; reference :12-20)."""
    classes = ops.pred_classes(preds)          # (H, N)
    maj, _ = torch.mode(classes, dim=0)
    return maj


class _Sub:
    """Dataset view over a point subset."""

    def __init__(self, preds, idxs):
        self.preds = preds[:, idxs]
        self.device = preds.device


def run_realisation(preds, pseudo_labels, pseudo_best, epsilon, pool,
                    budget, rng):
    N = preds.shape[1]
    idxs = torch.tensor(rng.sample(range(N), min(pool, N)))
    sub = _Sub(preds, idxs)
    labels = pseudo_labels[idxs]
    mp = ModelPicker(sub, epsilon=epsilon)
    first_correct = None
    for t in range(min(budget, len(idxs))):
        idx, q = mp.get_next_item_to_label()
        mp.add_label(idx, int(labels[idx]), q)
        best = int(mp.get_best_model_prediction())
        if best == pseudo_best:
            if first_correct is None:
                first_correct = t + 1
        else:
            first_correct = None
    return first_correct


def run_grid_search(task, data_dir, eps_grid, realisations, pool, budget,
                    threshold, seed=0):
    ds = Dataset(os.path.join(data_dir, task + ".pt"), device="cpu")
    pseudo = majority_vote_labels(ds.preds)
    pseudo_losses = ops.accuracy_losses(ops.pred_classes(ds.preds), pseudo)
    pseudo_best = int(pseudo_losses.argmin())

    results = {}
    for eps in eps_grid:
        rng = random.Random(seed)
        successes, fastest = 0, []
        for _ in range(realisations):
            t = run_realisation(ds.preds, pseudo, pseudo_best, eps, pool,
                                budget, rng)
            if t is not None:
                successes += 1
                fastest.append(t)
        rate = successes / realisations
        mean_t = sum(fastest) / len(fastest) if fastest else float("inf")
        results[eps] = (rate, mean_t)
        print(f"eps={eps:.2f}: success={rate:.2f} mean_t={mean_t:.1f}")

    # pick the fastest epsilon among those meeting the success threshold,
    # falling back to the highest success rate
    ok = {e: r for e, r in results.items() if r[0] >= threshold}
    chosen = min(ok, key=lambda e: ok[e][1]) if ok else \
        max(results, key=lambda e: results[e][0])
    return chosen, results


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--task", default=None,
                    help="one task; or use --missing to sweep all tasks "
                         "in --data-dir that best_epsilons.json lacks "
                         "(reference: launch_missing_modelselector.py)")
    ap.add_argument("--missing", action="store_true")
    ap.add_argument("--data-dir", default="data")
    ap.add_argument("--eps", default="0.35:0.49:0.01",
                    help="start:stop:step grid")
    ap.add_argument("--realisations", type=int, default=100)
    ap.add_argument("--pool", type=int, default=1000)
    ap.add_argument("--budget", type=int, default=100)
    ap.add_argument("--threshold", type=float, default=0.9)
    ap.add_argument("--out", default="best_epsilons.json")
    args = ap.parse_args()

    lo, hi, st = (float(x) for x in args.eps.split(":"))
    grid = [round(lo + i * st, 4) for i in range(int((hi - lo) / st) + 1)]

    if args.missing:
        have = {}
        if os.path.exists(args.out):
            with open(args.out) as f:
                have = json.load(f)
        tasks = sorted(f[:-3] for f in os.listdir(args.data_dir)
                       if f.endswith(".pt")
                       and not f.endswith("_labels.pt")
                       and f[:-3] not in have)
        if not tasks:
            print("no tasks missing from", args.out)
            return
    else:
        assert args.task, "--task or --missing required"
        tasks = [args.task]

    for task in tasks:
        chosen, results = run_grid_search(
            task, args.data_dir, grid, args.realisations, args.pool,
            args.budget, args.threshold)
        print(f"best epsilon for {task}: {chosen}")
        best = {}
        if os.path.exists(args.out):
            with open(args.out) as f:
                best = json.load(f)
        best[task] = chosen
        tmp = args.out + ".tmp"
        with open(tmp, "w") as f:
            json.dump(best, f, indent=2, sort_keys=True)
        os.replace(tmp, args.out)  # atomic under concurrency
        print("wrote", args.out)


if __name__ == "__main__":
    main()
