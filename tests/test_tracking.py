"""Tracking store: MLflow-SQLite schema fidelity + harness query patterns."""
import sqlite3

import pytest

from coda_amd import tracking


@pytest.fixture()
def db(tmp_path, monkeypatch):
    path = str(tmp_path / "coda.sqlite")
    monkeypatch.setattr(tracking, "_TRACKING_PATH", path)
    monkeypatch.setattr(tracking, "_EXPERIMENT", None)
    monkeypatch.setattr(tracking, "_RUN_STACK", [])
    return path


def test_nested_runs_and_raw_schema_query(db):
    """The paper-analysis SQL (tab1.py:28-51) must work against the raw DB."""
    tracking.set_experiment("cifar10_5592")
    with tracking.start_run(run_name="cifar10_5592-coda"):
        tracking.log_params({"method": "coda", "iters": 100})
        with tracking.start_run(nested=True,
                                run_name="cifar10_5592-coda-0"):
            tracking.log_param("seed", 0)
            for step in range(1, 101):
                tracking.log_metric("regret", 0.01 * step, step=step)
                tracking.log_metric("cumulative regret", 0.02 * step,
                                    step=step)
            tracking.log_param("stochastic", False)

    conn = sqlite3.connect(db)
    SQL = """
    SELECT  e.name AS task, rn.value AS run_name, m.value AS cr, m.step
    FROM    metrics m
    JOIN    runs r ON m.run_uuid = r.run_uuid
    JOIN    experiments e ON r.experiment_id = e.experiment_id
    JOIN    tags t_parent ON r.run_uuid = t_parent.run_uuid
           AND t_parent.key = 'mlflow.parentRunId'
    LEFT JOIN tags rn ON r.run_uuid = rn.run_uuid
           AND rn.key = 'mlflow.runName'
    WHERE   m.key = 'cumulative regret' AND m.step = 100
      AND   r.lifecycle_stage = 'active' AND e.lifecycle_stage = 'active'
    """
    rows = conn.execute(SQL).fetchall()
    conn.close()
    assert len(rows) == 1
    task, run_name, cr, step = rows[0]
    assert task == "cifar10_5592"
    assert run_name == "cifar10_5592-coda-0"
    assert abs(cr - 2.0) < 1e-9
    assert step == 100


def test_search_runs_and_resume_protocol(db):
    """The harness's skip-if-finished logic (main.py:136-158)."""
    tracking.set_experiment("taskA")
    with tracking.start_run(run_name="taskA-iid"):
        with tracking.start_run(nested=True, run_name="taskA-iid-0"):
            tracking.log_param("stochastic", True)

    runs = tracking.search_runs(experiment_names=["taskA"],
                                filter_string="tags.mlflow.runName = 'taskA-iid-0'",
                                max_results=1)
    assert len(runs) == 1
    assert runs.status.values[0] == "FINISHED"
    assert runs["params.stochastic"].values[0] == "True"

    # resuming the parent run by id keeps a single parent run
    run_id = tracking.search_runs(
        experiment_names=["taskA"],
        filter_string="tags.mlflow.runName = 'taskA-iid'").run_id.values[0]
    with tracking.start_run(run_id=run_id, run_name="taskA-iid"):
        with tracking.start_run(nested=True, run_name="taskA-iid-1"):
            tracking.log_param("seed", 1)
    all_parents = tracking.search_runs(experiment_names=["taskA"])
    parent_rows = [r for _, r in all_parents.iterrows()
                   if "tags.mlflow.parentRunId" not in all_parents.columns
                   or not isinstance(r.get("tags.mlflow.parentRunId"), str)]
    assert len(parent_rows) == 1


def test_failed_run_status(db):
    tracking.set_experiment("taskB")
    with pytest.raises(ValueError):
        with tracking.start_run(run_name="taskB-x"):
            raise ValueError("boom")
    runs = tracking.search_runs(experiment_names=["taskB"])
    assert runs.status.values[0] == "FAILED"


def test_metric_history(db):
    tracking.set_experiment("taskC")
    with tracking.start_run(run_name="r") as r:
        for s in (1, 2, 3):
            tracking.log_metric("m", float(s) * 0.5, step=s)
        rid = r.info.run_id
    hist = tracking.get_metric_history(rid, "m")
    assert hist == [(1, 0.5), (2, 1.0), (3, 1.5)]


def test_delete(db):
    tracking.set_experiment("taskD")
    with tracking.start_run(run_name="r"):
        tracking.log_metric("m", 1.0, step=1)
    rid = tracking.search_runs(experiment_names=["taskD"]).run_id.values[0]
    tracking.delete_run(rid)
    assert len(tracking.search_runs(experiment_names=["taskD"])) == 0


def test_concurrent_writers(db):
    """8 processes logging into one DB simultaneously (the task-parallel
    launcher's pattern on an 8-GPU node): WAL + busy timeout must keep
    every write."""
    import subprocess
    import sys

    code = """
import sys
sys.path.insert(0, {repo!r})
from coda_amd import tracking
tracking.set_tracking_uri("sqlite:///" + {db!r})
tracking.set_experiment("stress")
wid = int(sys.argv[1])
with tracking.start_run(run_name=f"w{{wid}}"):
    for s in range(1, 21):
        tracking.log_metric("m", wid + s * 0.01, step=s)
    tracking.log_param("worker", wid)
"""
    import os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    script = code.format(repo=repo, db=db)
    procs = [subprocess.Popen([sys.executable, "-c", script, str(i)])
             for i in range(8)]
    for p in procs:
        assert p.wait(timeout=400) == 0
    import sqlite3
    conn = sqlite3.connect(db)
    n_runs = conn.execute("SELECT COUNT(*) FROM runs").fetchone()[0]
    n_metrics = conn.execute(
        "SELECT COUNT(*) FROM metrics WHERE key='m'").fetchone()[0]
    statuses = [r[0] for r in conn.execute("SELECT status FROM runs")]
    conn.close()
    assert n_runs == 8
    assert n_metrics == 8 * 20
    assert all(s == "FINISHED" for s in statuses)


def test_concurrent_set_experiment_same_name(db):
    """Two processes racing to create the SAME experiment (the launcher
    runs taskX-iid and taskX-coda concurrently) must both succeed - this
    was a UNIQUE-constraint crash before INSERT OR IGNORE."""
    import subprocess
    import sys
    import os as _os

    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    code = (
        "import sys; sys.path.insert(0, {repo!r})\n"
        "from coda_amd import tracking\n"
        "tracking.set_tracking_uri('sqlite:///' + {db!r})\n"
        "tracking.set_experiment('racetask')\n"
        "with tracking.start_run(run_name='r' + sys.argv[1]):\n"
        "    tracking.log_metric('m', 1.0, step=1)\n"
    ).format(repo=repo, db=db)
    procs = [subprocess.Popen([sys.executable, "-c", code, str(i)])
             for i in range(6)]
    rcs = [p.wait(timeout=200) for p in procs]
    assert all(rc == 0 for rc in rcs), rcs
    import sqlite3
    conn = sqlite3.connect(db)
    n_exp = conn.execute(
        "SELECT COUNT(*) FROM experiments WHERE name='racetask'"
    ).fetchone()[0]
    n_runs = conn.execute("SELECT COUNT(*) FROM runs").fetchone()[0]
    conn.close()
    assert n_exp == 1 and n_runs == 6
