"""Generate a synthetic benchmark suite spanning the reference's task
shapes (2..182 classes, 1k..100k points, varying
model-pool quality), for end-to-end pipeline runs without the
(non-downloadable) 26-task data."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from coda_amd.datasets import write_synthetic_task  # noqa: E402

SUITE = [
    # name,                 H,  N,     C,  best, worst
    ("synth_binary",        12, 5000,  2,  0.92, 0.60),
    ("synth_cifar_like",    10, 10000, 10, 0.90, 0.55),
    ("synth_domainnet_like", 8, 20000, 126, 0.65, 0.30),
    ("synth_wilds_like",    16, 8000,  62, 0.75, 0.40),
    ("synth_glue_like",     6,  1000,  3,  0.85, 0.60),
    ("synth_wide_pool",     48, 4000,  10, 0.88, 0.50),
    # DomainNet-shaped at real scale: the largest
    # reference tasks are H=10, N~100-200k, C=126 DomainNet pairs
    # (reference paper/fig3.py:129-193 memory table)
    ("synth_domainnet_full", 10, 200000, 126, 0.65, 0.30),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--data-dir", default="data")
    args = ap.parse_args()
    for i, (name, H, N, C, best, worst) in enumerate(SUITE):
        p = write_synthetic_task(args.data_dir, name=name, H=H, N=N, C=C,
                                 seed=100 + i, best_acc=best,
                                 worst_acc=worst)
        print("wrote", p, f"H={H} N={N} C={C}")


if __name__ == "__main__":
    main()
