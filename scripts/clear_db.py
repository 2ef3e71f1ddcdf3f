"""Delete the tracking DB, or selected experiments/runs, with confirmation.

Reference parity: scripts/clear_db.py:26-87.

Usage:
    python scripts/clear_db.py --all [--yes]
    python scripts/clear_db.py --experiments taskA,taskB [--yes]
    python scripts/clear_db.py --runs <run_id>[,<run_id>...] [--yes]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coda_amd import tracking  # noqa: E402


def confirm(prompt, assume_yes):
    if assume_yes:
        return True
    resp = input(f"{prompt} [y/N] ").strip().lower()
    return resp == "y"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--all", action="store_true",
                   help="Delete the whole DB file")
    p.add_argument("--experiments", default=None,
                   help="Comma-separated experiment names to delete")
    p.add_argument("--runs", default=None,
                   help="Comma-separated run ids to delete")
    p.add_argument("--hard", action="store_true",
                   help="Physically remove rows instead of marking deleted")
    p.add_argument("--yes", action="store_true", help="Skip confirmation")
    args = p.parse_args()

    db = tracking.get_tracking_path()
    if args.all:
        if not os.path.exists(db):
            print("No DB at", db)
            return
        if confirm(f"Delete ENTIRE tracking DB {db}?", args.yes):
            os.remove(db)
            for suffix in ("-wal", "-shm"):
                if os.path.exists(db + suffix):
                    os.remove(db + suffix)
            print("Deleted", db)
        return

    if args.experiments:
        names = args.experiments.split(",")
        existing = dict((n, i) for i, n in tracking.list_experiments())
        for name in names:
            if name not in existing:
                print("No experiment named", name)
                continue
            if confirm(f"Delete experiment '{name}' and all its runs?",
                       args.yes):
                tracking.delete_experiment(existing[name], hard=args.hard)
                print("Deleted experiment", name)
        return

    if args.runs:
        for run_id in args.runs.split(","):
            if confirm(f"Delete run {run_id}?", args.yes):
                tracking.delete_run(run_id, hard=args.hard)
                print("Deleted run", run_id)
        return

    p.print_help()


if __name__ == "__main__":
    main()
