"""Flagship benchmark: acquisition steps/sec of the CODA engine.

Measures the full active-model-selection serving step (the reference's
CANONICAL no-prefilter acquisition - EIG over EVERY disagreeing
unlabeled point - -> oracle label -> Dirichlet posterior update ->
pi_hat refresh -> best-model P(best)) on the ImageNet-1k-scale config
named in BASELINE.json (config 2): H=128 candidate models x 50k points
per GPU x C=1000 classes, synthetic prediction tensors, fp32 compute
(the reference's compute dtype - its loader up-casts storage to fp32,
coda/datasets.py:14).

Scaling is WEAK along the CANDIDATE POOL: each GPU contributes 50k
unlabeled points (the model axis stores sharded - 128/world models per
rank - while EIG work shards by candidate; see the replicated-Beta
design in PARITY.md). One shared selection loop runs over the combined
pool, so whole-job steps/sec at fixed per-GPU work is the metric. The
candidate axis is the one that scales per-GPU-flat for this algorithm:
model-axis growth makes every candidate's acquisition intrinsically
more expensive (distinct-classes x 2H per candidate), which no
implementation can hold per-GPU-constant. The reference publishes no
absolute throughput numbers => vs_baseline = null.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N > 1 the driver launches it under torch.distributed.run with one rank per
GPU. Rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import random
import time

import torch

from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset
from coda_amd.options import LOSS_FNS
from coda_amd.parallel import init_from_env, get_comm

# Headline config (overridable via env for CPU smoke testing only).
# PREFILTER=0 is the reference's canonical acquisition (main.py:49, run
# name `coda-lr=0.01-mult=2.0-no-prefilter`, paper/tab1.py:60): EIG over
# EVERY disagreeing unlabeled point each step - ~50k candidates here.
H_MODELS = int(os.environ.get("CODA_BENCH_H", 128))
N_PER_GPU = int(os.environ.get("CODA_BENCH_N", 50_000))
C_CLASSES = int(os.environ.get("CODA_BENCH_C", 1000))
PREFILTER_N = int(os.environ.get("CODA_BENCH_PREFILTER", 0))
CHUNK = int(os.environ.get("CODA_BENCH_CHUNK", 256))
# storage dtype for the prediction pool (fp32 | bf16 | fp8); compute is
# always fp32 (coda_amd/datasets.py STORAGE_DTYPES)
STORAGE = os.environ.get("CODA_BENCH_STORAGE", "fp32")


def synth_preds(model_idxs, N, C, device, seed_base=1234,
                dtype=torch.float32):
    """Per-model deterministic synthetic predictions: rank-independent.

    Model h's tensor depends only on (seed_base + h), so a sharded run sees
    exactly the data the 1-GPU run sees for the same global model.
    `dtype` is the STORAGE dtype: each model's softmax converts on the fly,
    so the pool never exists at fp32 (512 GB at the 1M-point config).
    """
    out = torch.empty(len(model_idxs), N, C, device=device, dtype=dtype)
    g = torch.Generator(device=device)
    labels_g = torch.Generator(device=device)
    labels_g.manual_seed(seed_base - 1)
    labels = torch.randint(0, C, (N,), generator=labels_g, device=device)
    for i, h in enumerate(model_idxs):
        h = int(h)
        acc_g = torch.Generator().manual_seed(seed_base - 2 + 7919 * h)
        acc = 0.55 + 0.4 * torch.rand(1, generator=acc_g)
        g.manual_seed(seed_base + h)
        logits = torch.randn(N, C, generator=g, device=device)
        correct = torch.rand(N, generator=g, device=device) < float(acc)
        wrong = torch.randint(1, C, (N,), generator=g, device=device)
        target = torch.where(correct, labels, (labels + wrong) % C)
        logits.scatter_add_(1, target.unsqueeze(1),
                            torch.full((N, 1), 4.0, device=device))
        out[i] = torch.softmax(logits, dim=-1).to(dtype)
    return out, labels


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=800)
    ap.add_argument("--warmup", type=int, default=50)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    comm = init_from_env() if world > 1 else get_comm()
    if world > 1 and args.gpus != world:
        raise SystemExit(
            f"--gpus {args.gpus} does not match WORLD_SIZE={world}; "
            "launch one rank per GPU (torch.distributed.run "
            "--nproc-per-node N ... bench.py --gpus N)")
    n_gpus = world if world > 1 else args.gpus
    if n_gpus != 1 and world == 1:
        raise SystemExit(
            f"--gpus {n_gpus} requested but WORLD_SIZE is unset; "
            "multi-GPU runs go through torch.distributed.run")
    H_TOTAL = H_MODELS                # fixed pool of models
    N_TOTAL = N_PER_GPU * comm.world  # weak scaling: 50k points per GPU

    if torch.cuda.is_available():
        device = comm.device or torch.device("cuda", 0)
    else:
        device = torch.device("cpu")

    shard = (comm.rank, comm.world) if comm.is_distributed else None
    model_idxs = list(range(comm.rank, H_TOTAL, comm.world)) \
        if shard else list(range(H_TOTAL))

    from coda_amd.datasets import STORAGE_DTYPES
    preds, labels = synth_preds(model_idxs, N_TOTAL, C_CLASSES, device,
                                dtype=STORAGE_DTYPES[STORAGE])
    ds = Dataset.from_tensors(preds, labels, device, shard=None)
    ds.total_models = H_TOTAL
    ds.shard = shard
    oracle = Oracle(ds, LOSS_FNS["acc"])

    random.seed(0)
    torch.manual_seed(0)
    selector = CODA(ds, comm=comm, prefilter_n=PREFILTER_N,
                    chunk_size=CHUNK)

    def one_step():
        idx, q = selector.get_next_item_to_label()
        y = oracle(int(idx))
        selector.add_label(idx, y, q)
        selector.get_best_model_prediction()

    def sync():
        comm.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        one_step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    et = torch.tensor([elapsed], device=device if device.type == "cuda"
                      else torch.device("cpu"))
    if comm.is_distributed:
        torch.distributed.all_reduce(et, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(et[0])

    if comm.rank == 0:
        ms_per_step = 1000.0 * elapsed / args.steps
        value = args.steps / elapsed
        print(json.dumps({
            "metric": "acquisition_steps_per_sec",
            "value": value,
            "unit": "steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32" if STORAGE == "fp32" else f"fp32-compute/{STORAGE}-storage",
            "data": "synthetic",
            "config": {
                "model": "coda-eig",
                "task": "imagenet1k-scale",
                "H_models": H_TOTAL,
                "N_points": N_TOTAL,
                "C_classes": C_CLASSES,
                "prefilter_n": PREFILTER_N,
                "chunk_size": CHUNK,
                "global_batch": PREFILTER_N or N_TOTAL,
                "seq_len": C_CLASSES,
                "parallelism":
                    f"cand-shard{comm.world}" if comm.world > 1
                    else "single",
            },
        }))


if __name__ == "__main__":
    main()
