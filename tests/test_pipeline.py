"""End-to-end pipeline: launcher -> tracking DB -> aggregate -> analysis.

Exercises the task-parallel launcher (CPU slots), the idempotence
protocol, the aggregation script, and the raw-SQL analysis queries on
synthetic tasks - the L4/L5 layers of the framework.
"""
import json
import os
import sqlite3
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def pipeline_dir(tmp_path_factory):
    from coda_amd.datasets import write_synthetic_task
    d = tmp_path_factory.mktemp("pipeline")
    data = d / "data"
    for i, name in enumerate(["taskx", "tasky"]):
        write_synthetic_task(str(data), name=name, H=5, N=120, C=4, seed=i)
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    # the launcher invokes main.py relative to cwd
    for f in ("main.py", "bench.py"):
        (d / f).symlink_to(os.path.join(REPO, f))
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "launch_all_tasks.py"),
         "--pred-dir", "data",
         "--methods", "iid,coda-lr=0.01-mult=2.0-no-prefilter",
         "--seeds", "2", "--iters", "3", "--gpus", "0",
         "--max-concurrent", "2", "--polling-interval", "0.2"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "All jobs completed!" in r.stdout
    return d, env


def test_launcher_writes_all_runs(pipeline_dir):
    d, _ = pipeline_dir
    conn = sqlite3.connect(str(d / "coda.sqlite"))
    exps = [r[0] for r in conn.execute(
        "SELECT name FROM experiments WHERE lifecycle_stage='active'")]
    assert set(exps) >= {"taskx", "tasky"}
    # iid: 2 seeds per task (stochastic); coda: 1 seed (deterministic here)
    n_child = conn.execute(
        "SELECT COUNT(*) FROM tags WHERE key='mlflow.parentRunId'"
    ).fetchone()[0]
    assert n_child == 6, n_child
    conn.close()


def test_launcher_idempotent(pipeline_dir):
    d, env = pipeline_dir
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "launch_all_tasks.py"),
         "--pred-dir", "data",
         "--methods", "iid,coda-lr=0.01-mult=2.0-no-prefilter",
         "--seeds", "2", "--iters", "3", "--gpus", "0"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0
    assert "No jobs to run!" in r.stdout


def test_aggregate_and_analysis(pipeline_dir):
    d, env = pipeline_dir
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts",
                                      "aggregate_results.py")],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    conn = sqlite3.connect(str(d / "coda.sqlite"))
    n_mean = conn.execute(
        "SELECT COUNT(*) FROM metrics WHERE key='mean_regret'").fetchone()[0]
    assert n_mean > 0
    conn.close()

    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "tab1.py"),
         "--step", "3"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert "taskx" in r.stdout and "tasky" in r.stdout
    assert "Random Sampling" in r.stdout and "CODA (Ours)" in r.stdout

    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "fig1.py"),
         "--out", "fig1.png"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert (d / "fig1.png").exists()

    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "fig5.py"),
         "--out", "fig5.png"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert (d / "fig5.png").exists()

    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "fig4.py")],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert "failure rate" in r.stdout

    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "fig3.py"),
         "--data-dir", "data"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert "by class count" in r.stdout


def test_clear_db_selected(pipeline_dir):
    d, env = pipeline_dir
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "clear_db.py"),
         "--experiments", "tasky", "--yes"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=200)
    assert r.returncode == 0, r.stderr
    conn = sqlite3.connect(str(d / "coda.sqlite"))
    exps = [x[0] for x in conn.execute(
        "SELECT name FROM experiments WHERE lifecycle_stage='active'")]
    assert "tasky" not in exps and "taskx" in exps
    conn.close()


def test_eps_gridsearch(pipeline_dir):
    d, env = pipeline_dir
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts",
                                      "modelpicker_eps_gridsearch.py"),
         "--task", "taskx", "--data-dir", "data",
         "--eps", "0.40:0.48:0.04", "--realisations", "5",
         "--pool", "60", "--budget", "15"],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    best = json.loads((d / "best_epsilons.json").read_text())
    assert "taskx" in best


def test_latency_report(pipeline_dir):
    d, env = pipeline_dir
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "analysis", "latency.py")],
        cwd=str(d), env=env, capture_output=True, text=True, timeout=400)
    assert r.returncode == 0, r.stderr
    assert "steps/s" in r.stdout and "taskx" in r.stdout
