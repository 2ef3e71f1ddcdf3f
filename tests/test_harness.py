"""End-to-end harness: main.py CLI on a synthetic task (CPU)."""
import os
import sqlite3

import pytest
import torch

import main as harness
from coda_amd.datasets import write_synthetic_task
from coda_amd import tracking


@pytest.fixture()
def task_dir(tmp_path):
    write_synthetic_task(str(tmp_path / "data"), name="synthtask",
                         H=6, N=200, C=4, seed=0)
    return tmp_path


def _run_cli(tmp_path, extra):
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        tracking.set_tracking_uri("sqlite:///coda.sqlite")
        tracking._EXPERIMENT = None
        tracking._RUN_STACK.clear()
        harness.main(["--task", "synthtask", "--data-dir", "data",
                      "--device", "cpu"] + extra)
    finally:
        os.chdir(cwd)


def test_no_mlflow_smoke(task_dir):
    _run_cli(task_dir, ["--method", "coda", "--iters", "3", "--seeds", "1",
                        "--no-mlflow", "--chunk-size", "64"])
    assert not (task_dir / "coda.sqlite").exists()


def test_cli_writes_schema_and_skips_finished(task_dir, capsys):
    _run_cli(task_dir, ["--method", "iid", "--iters", "5", "--seeds", "2"])
    db = task_dir / "coda.sqlite"
    assert db.exists()
    conn = sqlite3.connect(str(db))
    n_child = conn.execute(
        "SELECT COUNT(*) FROM tags WHERE key='mlflow.parentRunId'"
    ).fetchone()[0]
    assert n_child == 2  # iid is stochastic -> both seeds ran
    n_metrics = conn.execute(
        "SELECT COUNT(*) FROM metrics WHERE key='regret'").fetchone()[0]
    assert n_metrics == 10  # 5 iters x 2 seeds
    conn.close()

    # re-run: both seeds finished -> skipped
    _run_cli(task_dir, ["--method", "iid", "--iters", "5", "--seeds", "2"])
    out = capsys.readouterr().out
    assert "Skipping" in out


def test_deterministic_method_stops_after_seed0(task_dir, capsys):
    # CODA with no EIG ties is deterministic -> the harness must stop after
    # seed 0 (main.py:164-168 in the reference). uncertainty/iid are
    # effectively always stochastic (risk ties at step 0).
    _run_cli(task_dir, ["--method", "coda", "--iters", "2",
                        "--seeds", "3", "--chunk-size", "64"])
    conn = sqlite3.connect(str(task_dir / "coda.sqlite"))
    n_child = conn.execute(
        "SELECT COUNT(*) FROM tags WHERE key='mlflow.parentRunId'"
    ).fetchone()[0]
    conn.close()
    assert n_child == 1  # deterministic -> only seed 0


@pytest.mark.parametrize("method", ["coda", "activetesting", "vma",
                                    "model_picker", "uncertainty"])
def test_all_methods_run_one_iter(task_dir, method):
    _run_cli(task_dir, ["--method", method, "--iters", "1", "--seeds", "1",
                        "--no-mlflow", "--chunk-size", "64"])


def test_checkpoint_resume(task_dir, monkeypatch):
    """A seed killed mid-run resumes from its checkpoint and produces the
    SAME metrics as an uninterrupted run (RNG state checkpointed too)."""
    import sqlite3 as sq
    import main as harness_mod

    def regrets(exp):
        conn = sq.connect(str(task_dir / "coda.sqlite"))
        rows = conn.execute(
            "SELECT m.step, m.value FROM metrics m "
            "JOIN runs r ON m.run_uuid=r.run_uuid "
            "JOIN experiments e ON r.experiment_id=e.experiment_id "
            "WHERE e.name=? AND m.key='regret' ORDER BY m.step",
            (exp,)).fetchall()
        conn.close()
        return rows

    # uninterrupted reference run
    _run_cli(task_dir, ["--method", "iid", "--iters", "6", "--seeds", "1",
                        "--experiment-name", "full"])
    want = regrets("full")
    assert len(want) == 6

    # crash after step 3: patch the oracle to explode on the 4th call
    calls = {"n": 0}
    orig = harness_mod.Oracle.__call__

    def exploding(self, idx):
        calls["n"] += 1
        if calls["n"] > 3:
            raise RuntimeError("simulated crash")
        return orig(self, idx)

    monkeypatch.setattr(harness_mod.Oracle, "__call__", exploding)
    try:
        _run_cli(task_dir, ["--method", "iid", "--iters", "6", "--seeds",
                            "1", "--experiment-name", "ck",
                            "--checkpoint-every", "1",
                            "--checkpoint-dir", str(task_dir / "ck")])
    except RuntimeError:
        pass
    monkeypatch.setattr(harness_mod.Oracle, "__call__", orig)
    assert list((task_dir / "ck").glob("*.pt")), "no checkpoint left"

    # resume: run id exists with FAILED status -> re-runs, resuming at 3
    _run_cli(task_dir, ["--method", "iid", "--iters", "6", "--seeds", "1",
                        "--experiment-name", "ck", "--checkpoint-every",
                        "1", "--checkpoint-dir", str(task_dir / "ck")])
    got = regrets("ck")
    assert got == want, (got, want)
    assert not list((task_dir / "ck").glob("*.pt")), "checkpoint not cleaned"


def test_checkpoint_state_roundtrip(task_dir):
    """ckpt.state_dict/load_state_dict restores CODA exactly."""
    import random as rnd
    import torch
    from coda_amd import CODA, Oracle, checkpoint as ckpt
    from coda_amd.datasets import Dataset
    from coda_amd.options import LOSS_FNS

    path = os.path.join(str(task_dir), "data", "synthtask.pt")
    ds = Dataset(path, "cpu")
    ds.labels = torch.load(path.replace(".pt", "_labels.pt"),
                           weights_only=True)
    oracle = Oracle(ds, LOSS_FNS["acc"])
    rnd.seed(0); torch.manual_seed(0)
    sel = CODA(ds, chunk_size=64)
    for _ in range(3):
        i, q = sel.get_next_item_to_label()
        sel.add_label(i, oracle(int(i)), q)
    state = ckpt.state_dict(sel)

    rnd.seed(0); torch.manual_seed(0)
    sel2 = CODA(ds, chunk_size=64)
    ckpt.load_state_dict(sel2, state)
    assert sel2.labeled_idxs == sel.labeled_idxs
    assert sel2.step == sel.step
    torch.testing.assert_close(sel2.dirichlets, sel.dirichlets)
    torch.testing.assert_close(sel2.pi_hat, sel.pi_hat)
    # both continue identically
    rnd.seed(42); torch.manual_seed(42)
    i1, q1 = sel.get_next_item_to_label()
    rnd.seed(42); torch.manual_seed(42)
    i2, q2 = sel2.get_next_item_to_label()
    assert int(i1) == int(i2) and abs(q1 - q2) < 1e-6


def test_model_picker_reads_tuned_epsilon(task_dir, capsys):
    """best_epsilons.json (from the grid search) overrides TASK_EPS."""
    import json
    import main as harness_mod
    (task_dir / "best_epsilons.json").write_text(
        json.dumps({"synthtask": 0.41}))
    cwd = os.getcwd()
    os.chdir(task_dir)
    try:
        args = harness_mod.parse_args(
            ["--task", "synthtask", "--data-dir", "data",
             "--method", "model_picker", "--device", "cpu"])
        from coda_amd.datasets import Dataset
        ds = Dataset(str(task_dir / "data" / "synthtask.pt"), "cpu")
        from coda_amd.options import LOSS_FNS
        sel = harness_mod.build_selector(ds, args, LOSS_FNS["acc"])
        assert abs(sel.epsilon - 0.41) < 1e-9
    finally:
        os.chdir(cwd)


def test_convert_task_roundtrip(task_dir):
    import subprocess, sys
    import torch
    r = subprocess.run(
        [sys.executable,
         os.path.join(os.path.dirname(os.path.dirname(
             os.path.abspath(__file__))), "scripts", "convert_task.py"),
         "--task", "synthtask", "--data-dir", str(task_dir / "data"),
         "--dtype", "bf16", "--out-dir", str(task_dir / "data_bf16")],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    from coda_amd.datasets import Dataset
    a = Dataset(str(task_dir / "data" / "synthtask.pt"), "cpu")
    b = Dataset(str(task_dir / "data_bf16" / "synthtask.pt"), "cpu")
    assert b.preds.dtype == torch.float32  # up-cast on load
    torch.testing.assert_close(a.preds, b.preds, rtol=1e-2, atol=5e-3)
    assert b.labels is not None


def test_slashed_task_names(tmp_path):
    """GLUE-style task names contain '/' (TASK_EPS: 'glue/cola'); the
    harness, tracking DB and checkpoints must all handle them."""
    from coda_amd.datasets import write_synthetic_task
    import sqlite3 as sq
    write_synthetic_task(str(tmp_path / "data" / "glue"), name="cola",
                         H=4, N=80, C=3, seed=5)
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        from coda_amd import tracking
        tracking.set_tracking_uri("sqlite:///coda.sqlite")
        tracking._EXPERIMENT = None
        tracking._RUN_STACK.clear()
        import main as harness
        harness.main(["--task", "glue/cola", "--data-dir", "data",
                      "--device", "cpu", "--method", "iid", "--iters", "2",
                      "--seeds", "1", "--checkpoint-every", "1",
                      "--checkpoint-dir", "ck"])
    finally:
        os.chdir(cwd)
    conn = sq.connect(str(tmp_path / "coda.sqlite"))
    exps = [r[0] for r in conn.execute("SELECT name FROM experiments")]
    n = conn.execute("SELECT COUNT(*) FROM metrics WHERE key='regret'"
                     ).fetchone()[0]
    conn.close()
    assert "glue/cola" in exps and n == 2


def test_storage_flag_cpu(task_dir):
    """--storage bf16 runs the harness end-to-end with a bf16 pool on CPU
    (compute upcasts per chunk)."""
    _run_cli(task_dir, ["--method", "coda", "--iters", "2", "--seeds", "1",
                        "--no-mlflow", "--chunk-size", "64",
                        "--storage", "bf16"])


def test_bench_rejects_gpu_count_mismatch():
    """bench.py must fail loudly when --gpus disagrees with the launch
    (single-process with --gpus 8 printed a fake rank count in r01)."""
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run([sys.executable, "bench.py", "--gpus", "8",
                        "--steps", "1", "--warmup", "0"],
                       cwd=repo, env=env, capture_output=True, text=True,
                       timeout=120)
    assert r.returncode != 0
    assert "WORLD_SIZE" in (r.stdout + r.stderr)
