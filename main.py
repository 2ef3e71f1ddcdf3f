"""Experiment harness CLI - `python main.py --task T --method M [...]`.

CLI-flag and behavior parity with the reference harness (main.py:28-168):
same flags, same seed handling (deterministic methods stop after seed 0 via
the selector's `stochastic` attribute), same skip-if-finished resume, same
MLflow-SQLite result schema (experiment = task, parent run = method, child
run = seed; metrics "regret" and "cumulative regret" at steps 1..iters).

Tracking goes through coda_amd.tracking, a native store writing the raw
MLflow SQLite schema into coda.sqlite.

New MI355X flags (additive; defaults reproduce the reference behavior):
  --device         cpu / cuda (default: cuda if available)
  --chunk-size     EIG candidate chunk (reference hard-codes 100)
  --sharded        shard the model axis across torchrun ranks (RCCL/xGMI)
"""
from __future__ import annotations

import argparse
import os
import random
import time

import numpy as np
import torch
from tqdm import tqdm

from coda_amd import CODA, Dataset, Oracle
from coda_amd.baselines import (IID, ActiveTesting, VMA, ModelPicker,
                                Uncertainty, TASK_EPS)
from coda_amd.options import LOSS_FNS
from coda_amd import tracking
from coda_amd.parallel import init_from_env, get_comm

tracking.set_tracking_uri("sqlite:///coda.sqlite")


def seed_all(seed: int):
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
        torch.cuda.manual_seed_all(seed)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False


def parse_args(argv=None):
    parser = argparse.ArgumentParser()
    # dataset settings
    parser.add_argument("--task", default=None,
                        help="{ 'sketch_painting', ... }")
    parser.add_argument("--data-dir", default="data")

    # benchmarking settings
    parser.add_argument("--iters", type=int, default=100)
    parser.add_argument("--seeds", type=int, default=5)
    parser.add_argument("--force-rerun", action="store_true",
                        help="Overwrite existing runs.")
    parser.add_argument("--experiment-name", default=None)
    parser.add_argument("--no-mlflow", action="store_true",
                        help="Disable result-DB logging.")

    # general method settings
    parser.add_argument("--loss", default="acc")
    parser.add_argument("--method", default="iid",
                        help="{ 'iid', 'coda*', 'uncertainty', "
                             "'activetesting', 'vma', 'model_picker' }")

    # CODA settings
    parser.add_argument("--alpha", default=0.9, type=float)
    parser.add_argument("--learning-rate", default=0.01, type=float)
    parser.add_argument("--multiplier", default=2.0, type=float)
    parser.add_argument("--prefilter-n", type=int, default=0)
    parser.add_argument("--no-diag-prior", action="store_true")
    parser.add_argument("--q", default="eig",
                        help="Acquisition {eig, iid, uncertainty}.")

    # MI355X additions
    parser.add_argument("--device", default=None, help="cpu / cuda")
    parser.add_argument("--chunk-size", type=int, default=100)
    parser.add_argument("--pi-hat-precision", default="auto",
                        choices=["auto", "fp32", "bf16"],
                        help="pi_hat contraction dtype. 'auto' is the "
                             "documented bf16 MFMA fast path on GPU "
                             "(fp32 accumulate); pass fp32 for strict "
                             "bit-parity with the reference's fp32 GEMM.")
    parser.add_argument("--debug-checks", action="store_true",
                        help="Enable the per-stage NaN/Inf and "
                             "probability guards (the reference's "
                             "default-on _DEBUG behavior; each check "
                             "synchronizes the device).")
    parser.add_argument("--sharded", action="store_true",
                        help="Shard the model axis across torchrun ranks.")
    parser.add_argument("--storage", default="fp32",
                        choices=["fp32", "bf16", "fp8"],
                        help="On-device prediction-pool dtype (compute "
                             "always upcasts to fp32).")
    parser.add_argument("--stream-chunk-mb", type=int, default=0,
                        help="Stream the pool host->HBM through pinned "
                             "buffers of this size (0 = one blocking copy).")
    parser.add_argument("--checkpoint-every", type=int, default=0,
                        help="Save selector state every K steps (0 = off); "
                             "an interrupted seed resumes mid-run.")
    parser.add_argument("--checkpoint-dir", default="checkpoints")
    return parser.parse_args(argv)


def build_selector(dataset, args, loss_fn, comm=None):
    if (comm is not None and comm.is_distributed
            and not args.method.startswith("coda")):
        # Baselines are shard-unaware: their best-model index would be
        # shard-LOCAL while regret is evaluated in GLOBAL model order.
        raise ValueError(
            f"--sharded supports only CODA methods; '{args.method}' "
            "would compute regret against the wrong model. Run "
            "baselines single-process (the task-parallel launcher "
            "covers multi-GPU baseline sweeps).")
    if args.method == "iid":
        return IID(dataset, loss_fn)
    if args.method == "uncertainty":
        return Uncertainty(dataset, loss_fn)
    if args.method.startswith("coda"):
        return CODA.from_args(dataset, args, comm=comm)
    if args.method == "activetesting":
        return ActiveTesting(dataset, loss_fn)
    if args.method == "vma":
        return VMA(dataset, loss_fn)
    if args.method == "model_picker":
        # per-task epsilon: the built-in table, overridden by a tuned
        # best_epsilons.json (scripts/modelpicker_eps_gridsearch.py)
        eps = TASK_EPS.get(args.task)
        import json
        for cand in ("best_epsilons.json",
                     os.path.join(args.data_dir, "best_epsilons.json")):
            if os.path.exists(cand):
                try:
                    tuned = json.load(open(cand))
                    if args.task in tuned:
                        eps = float(tuned[args.task])
                except (ValueError, OSError):
                    pass
        if eps is None:
            print(args.task, "not in TASK_EPS; using default")
            return ModelPicker(dataset)
        return ModelPicker(dataset, epsilon=eps)
    raise ValueError(args.method + " is not a supported method.")


def do_model_selection_experiment(dataset, oracle, args, loss_fn, seed=0,
                                  comm=None, log=True):
    from coda_amd import checkpoint as ckpt
    comm = comm or get_comm()
    seed_all(seed)
    true_losses = oracle.true_losses(dataset.preds)
    if comm.is_distributed:
        # per-model losses live on shards; gather to global order
        gathered = comm.all_gather_cat(true_losses, dim=0)
        true_losses = gathered[
            comm.unshard_order(dataset.total_models).to(gathered.device)]
    best_loss = true_losses.min()
    is_main = comm.rank == 0
    if is_main:
        print("Best possible loss is", float(best_loss))

    selector = build_selector(dataset, args, loss_fn, comm=comm)

    best_model_idx_pred = selector.get_best_model_prediction()
    regret_loss = true_losses[best_model_idx_pred] - best_loss
    if is_main:
        print("Regret at 0:", float(regret_loss))

    # mid-run checkpoint/resume (new capability)
    ckpt_path = None
    start_m, cumulative_regret_loss = 0, 0.0
    if args.checkpoint_every:
        os.makedirs(args.checkpoint_dir, exist_ok=True)
        safe_task = args.task.replace("/", "_")  # GLUE tasks: 'glue/cola'
        ckpt_path = os.path.join(
            args.checkpoint_dir, f"{safe_task}-{args.method}-{seed}.pt")
        if ckpt.exists(ckpt_path, comm.rank, comm.world):
            blob = torch.load(
                ckpt_path + (f".rank{comm.rank}" if comm.world > 1 else ""),
                map_location="cpu", weights_only=False)
            ckpt.load_state_dict(selector, blob["selector"])
            start_m = blob["m"]
            cumulative_regret_loss = blob["cumulative"]
            random.setstate(blob["py_rng"])
            torch.set_rng_state(blob["torch_rng"])
            if is_main:
                print(f"Resumed {ckpt_path} at step {start_m}")

    iterator = tqdm(range(start_m, args.iters)) if is_main \
        else range(start_m, args.iters)
    for m in iterator:
        t0 = time.perf_counter()
        chosen_idx, selection_prob = selector.get_next_item_to_label()
        true_class = oracle(chosen_idx)
        selector.add_label(chosen_idx, true_class, selection_prob)
        best_model_idx_pred = selector.get_best_model_prediction()
        step_s = time.perf_counter() - t0

        regret_loss = true_losses[best_model_idx_pred] - best_loss
        cumulative_regret_loss += float(regret_loss)
        if log and is_main and not args.no_mlflow:
            tracking.log_metric("regret", float(regret_loss), step=m + 1)
            tracking.log_metric("cumulative regret",
                                float(cumulative_regret_loss), step=m + 1)
            tracking.log_metric("step_seconds", step_s, step=m + 1)

        if ckpt_path and (m + 1) % args.checkpoint_every == 0:
            blob = {"selector": ckpt.state_dict(selector), "m": m + 1,
                    "cumulative": cumulative_regret_loss,
                    "py_rng": random.getstate(),
                    "torch_rng": torch.get_rng_state()}
            path = ckpt_path + (f".rank{comm.rank}" if comm.world > 1
                                else "")
            torch.save(blob, path + ".tmp")
            os.replace(path + ".tmp", path)

    if ckpt_path:
        path = ckpt_path + (f".rank{comm.rank}" if comm.world > 1 else "")
        if os.path.exists(path):
            os.remove(path)

    return selector.stochastic


def main(argv=None):
    args = parse_args(argv)

    if args.debug_checks:
        from coda_amd.util import set_debug
        set_debug(True)

    comm = init_from_env() if args.sharded else get_comm()
    if args.device is not None:
        device = torch.device(args.device)
    elif comm.device is not None:
        device = comm.device
    else:
        device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    is_main = comm.rank == 0
    if is_main:
        print("device is", device)

    shard = (comm.rank, comm.world) if comm.is_distributed else None
    dataset = Dataset(os.path.join(args.data_dir, args.task + ".pt"),
                      device=device, shard=shard,
                      storage_dtype=args.storage,
                      stream_chunk_mb=args.stream_chunk_mb)
    loss_fn = LOSS_FNS[args.loss]
    oracle = Oracle(dataset, loss_fn=loss_fn)

    if args.no_mlflow or not is_main:
        for seed in range(args.seeds):
            if is_main:
                print("Running active model selection with seed", seed)
            stoch = do_model_selection_experiment(
                dataset, oracle, args, loss_fn, seed=seed, comm=comm,
                log=False)
            if not stoch:
                if is_main:
                    print("Method is not stochastic for this task. "
                          "Skipping further seeds.")
                break
        return

    experiment_name = args.experiment_name or args.task
    tracking.set_experiment(experiment_name)

    def get_run_id(run_name):
        runs = tracking.search_runs(
            experiment_names=[experiment_name],
            filter_string=f"tags.mlflow.runName = '{run_name}'",
            max_results=1)
        if len(runs) == 0:
            return None, False, None
        run_id = runs.run_id.values[0]
        finished = runs.status.values[0] == "FINISHED"
        stoch = ("params.stochastic" in runs.columns and
                 runs["params.stochastic"].values[0] == "True")
        return run_id, finished, stoch

    run_name = "-".join([experiment_name, args.method])
    run_id, _, _ = get_run_id(run_name)
    with tracking.start_run(run_id=run_id, run_name=run_name):
        tracking.log_params(args.__dict__)
        for seed in range(args.seeds):
            seed_run_name = "-".join([experiment_name, args.method, str(seed)])
            seed_run_id, seed_finished, seed_stochastic = get_run_id(
                seed_run_name)
            if seed_finished and not args.force_rerun:
                print("Seed", seed, "finished. Skipping.")
            else:
                with tracking.start_run(nested=True, run_id=seed_run_id,
                                        run_name=seed_run_name):
                    tracking.log_param("seed", seed)
                    print("Running active model selection with seed", seed)
                    seed_stochastic = do_model_selection_experiment(
                        dataset, oracle, args, loss_fn, seed=seed, comm=comm)
                    tracking.log_param("stochastic", seed_stochastic)

            if not seed_stochastic:
                print("Method is not stochastic for this task. "
                      "Skipping further seeds.")
                break


if __name__ == "__main__":
    main()
