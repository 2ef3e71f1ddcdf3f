"""Native experiment-tracking store, MLflow-SQLite-schema compatible.

The reference logs through MLflow to `sqlite:///coda.sqlite` (main.py:15-17)
and its analysis layer queries the RAW MLflow tables
(metrics JOIN runs JOIN experiments JOIN tags with 'mlflow.parentRunId' /
'mlflow.runName' tags - paper/tab1.py:28-51), so the on-disk layout, not
just the API, is the contract. This module implements that contract
directly on sqlite3 - same tables, same columns, same tag keys, same
3-level hierarchy (experiment = task, parent run = method, child run =
seed) - with the subset of the `mlflow` module API the harness and
scripts use:

    set_tracking_uri, set_experiment, start_run (context manager, nested),
    log_param(s), log_metric, log_image, search_runs -> pandas DataFrame.

If a real `mlflow` package queries the produced .sqlite file, the standard
joins work; conversely this module can read DBs produced by MLflow 2.x.

Multi-process safety (the local task-parallel launcher runs up to 8
harness processes against one DB): WAL journal + busy timeout + one
transaction per logging call.
"""
from __future__ import annotations

import contextlib
import os
import time
import uuid
from typing import Optional

import sqlite3

_TRACKING_PATH = "coda.sqlite"
_EXPERIMENT: Optional[dict] = None
_RUN_STACK: list = []


# ---------------------------------------------------------------------------
# Schema
# ---------------------------------------------------------------------------

_SCHEMA = """
CREATE TABLE IF NOT EXISTS experiments (
    experiment_id INTEGER PRIMARY KEY AUTOINCREMENT,
    name VARCHAR(256) UNIQUE,
    artifact_location VARCHAR(256),
    lifecycle_stage VARCHAR(32) DEFAULT 'active',
    creation_time BIGINT,
    last_update_time BIGINT
);
CREATE TABLE IF NOT EXISTS runs (
    run_uuid VARCHAR(32) PRIMARY KEY,
    name VARCHAR(250),
    source_type VARCHAR(20),
    source_name VARCHAR(500),
    entry_point_name VARCHAR(50),
    user_id VARCHAR(256),
    status VARCHAR(9),
    start_time BIGINT,
    end_time BIGINT,
    source_version VARCHAR(50),
    lifecycle_stage VARCHAR(20) DEFAULT 'active',
    artifact_uri VARCHAR(200),
    experiment_id INTEGER REFERENCES experiments(experiment_id),
    deleted_time BIGINT
);
CREATE TABLE IF NOT EXISTS metrics (
    key VARCHAR(250) NOT NULL,
    value FLOAT NOT NULL,
    timestamp BIGINT NOT NULL,
    run_uuid VARCHAR(32) NOT NULL REFERENCES runs(run_uuid),
    step BIGINT NOT NULL DEFAULT 0,
    is_nan BOOLEAN NOT NULL DEFAULT 0,
    PRIMARY KEY (key, timestamp, step, run_uuid, value, is_nan)
);
CREATE TABLE IF NOT EXISTS latest_metrics (
    key VARCHAR(250) NOT NULL,
    value FLOAT NOT NULL,
    timestamp BIGINT,
    step BIGINT NOT NULL,
    is_nan BOOLEAN NOT NULL,
    run_uuid VARCHAR(32) NOT NULL REFERENCES runs(run_uuid),
    PRIMARY KEY (key, run_uuid)
);
CREATE TABLE IF NOT EXISTS params (
    key VARCHAR(250) NOT NULL,
    value VARCHAR(8000) NOT NULL,
    run_uuid VARCHAR(32) NOT NULL REFERENCES runs(run_uuid),
    PRIMARY KEY (key, run_uuid)
);
CREATE TABLE IF NOT EXISTS tags (
    key VARCHAR(250) NOT NULL,
    value VARCHAR(8000),
    run_uuid VARCHAR(32) NOT NULL REFERENCES runs(run_uuid),
    PRIMARY KEY (key, run_uuid)
);
CREATE INDEX IF NOT EXISTS index_metrics_run_uuid ON metrics (run_uuid);
CREATE INDEX IF NOT EXISTS index_params_run_uuid ON params (run_uuid);
CREATE INDEX IF NOT EXISTS index_tags_run_uuid ON tags (run_uuid);
"""


def _connect() -> sqlite3.Connection:
    conn = sqlite3.connect(_TRACKING_PATH, timeout=60.0)
    conn.execute("PRAGMA journal_mode=WAL")
    conn.execute("PRAGMA busy_timeout=60000")
    return conn


def _ensure_schema(conn: sqlite3.Connection):
    conn.executescript(_SCHEMA)
    conn.commit()


def _now_ms() -> int:
    return int(time.time() * 1000)


# ---------------------------------------------------------------------------
# Public API (mlflow-compatible subset)
# ---------------------------------------------------------------------------

def set_tracking_uri(uri: str):
    """Accepts 'sqlite:///path.sqlite' or a bare filesystem path."""
    global _TRACKING_PATH
    if uri.startswith("sqlite:///"):
        _TRACKING_PATH = uri[len("sqlite:///"):]
    else:
        _TRACKING_PATH = uri


def get_tracking_path() -> str:
    return _TRACKING_PATH


def set_experiment(name: str) -> dict:
    global _EXPERIMENT
    conn = _connect()
    try:
        _ensure_schema(conn)
        now = _now_ms()
        # race-safe against concurrent harness processes creating the same
        # experiment (the task-parallel launcher runs task x method jobs
        # concurrently): INSERT OR IGNORE, then read back
        conn.execute(
            "INSERT OR IGNORE INTO experiments (name, artifact_location, "
            "lifecycle_stage, creation_time, last_update_time) "
            "VALUES (?, ?, 'active', ?, ?)",
            (name, f"./mlruns/{name}", now, now))
        conn.commit()
        exp_id = conn.execute(
            "SELECT experiment_id FROM experiments WHERE name = ?",
            (name,)).fetchone()[0]
        _EXPERIMENT = {"experiment_id": exp_id, "name": name}
        return _EXPERIMENT
    finally:
        conn.close()


class ActiveRun:
    def __init__(self, run_id: str, run_name: str):
        self.info = type("RunInfo", (), {"run_id": run_id,
                                         "run_name": run_name})()

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc, tb):
        end_run("FAILED" if exc_type else "FINISHED")
        return False


def start_run(run_id: Optional[str] = None, run_name: Optional[str] = None,
              nested: bool = False) -> ActiveRun:
    if _EXPERIMENT is None:
        set_experiment("Default")
    if _RUN_STACK and not nested:
        raise RuntimeError("Run already active; pass nested=True")
    parent_id = _RUN_STACK[-1] if _RUN_STACK else None

    conn = _connect()
    try:
        _ensure_schema(conn)
        if run_id is not None:
            row = conn.execute(
                "SELECT run_uuid, name FROM runs WHERE run_uuid = ?",
                (run_id,)).fetchone()
            if row is not None:
                run_name = run_name or row[1]
                conn.execute(
                    "UPDATE runs SET status='RUNNING', end_time=NULL "
                    "WHERE run_uuid = ?", (run_id,))
                conn.commit()
                _RUN_STACK.append(run_id)
                return ActiveRun(run_id, run_name)
        new_id = run_id or uuid.uuid4().hex
        run_name = run_name or f"run-{new_id[:8]}"
        now = _now_ms()
        conn.execute(
            "INSERT INTO runs (run_uuid, name, source_type, source_name, "
            "entry_point_name, user_id, status, start_time, end_time, "
            "source_version, lifecycle_stage, artifact_uri, experiment_id) "
            "VALUES (?, ?, 'LOCAL', '', '', ?, 'RUNNING', ?, NULL, '', "
            "'active', ?, ?)",
            (new_id, run_name, os.environ.get("USER", "coda"), now,
             f"./mlruns/{_EXPERIMENT['name']}/{new_id}/artifacts",
             _EXPERIMENT["experiment_id"]))
        conn.execute(
            "INSERT OR REPLACE INTO tags (key, value, run_uuid) VALUES "
            "('mlflow.runName', ?, ?)", (run_name, new_id))
        if parent_id is not None:
            conn.execute(
                "INSERT OR REPLACE INTO tags (key, value, run_uuid) VALUES "
                "('mlflow.parentRunId', ?, ?)", (parent_id, new_id))
        conn.commit()
        _RUN_STACK.append(new_id)
        return ActiveRun(new_id, run_name)
    finally:
        conn.close()


def end_run(status: str = "FINISHED"):
    if not _RUN_STACK:
        return
    run_id = _RUN_STACK.pop()
    conn = _connect()
    try:
        conn.execute("UPDATE runs SET status=?, end_time=? WHERE run_uuid=?",
                     (status, _now_ms(), run_id))
        conn.commit()
    finally:
        conn.close()


def active_run_id() -> Optional[str]:
    return _RUN_STACK[-1] if _RUN_STACK else None


def log_param(key: str, value):
    _log_params_impl({key: value})


def log_params(params: dict):
    _log_params_impl(params)


def _log_params_impl(params: dict):
    run_id = active_run_id()
    if run_id is None:
        raise RuntimeError("No active run")
    conn = _connect()
    try:
        conn.executemany(
            "INSERT OR REPLACE INTO params (key, value, run_uuid) "
            "VALUES (?, ?, ?)",
            [(k, str(v), run_id) for k, v in params.items()])
        conn.commit()
    finally:
        conn.close()


def log_metric(key: str, value: float, step: int = 0):
    run_id = active_run_id()
    if run_id is None:
        raise RuntimeError("No active run")
    value = float(value)
    is_nan = int(value != value)
    now = _now_ms()
    conn = _connect()
    try:
        conn.execute(
            "INSERT OR REPLACE INTO metrics "
            "(key, value, timestamp, run_uuid, step, is_nan) "
            "VALUES (?, ?, ?, ?, ?, ?)",
            (key, value, now, run_id, step, is_nan))
        conn.execute(
            "INSERT OR REPLACE INTO latest_metrics "
            "(key, value, timestamp, step, is_nan, run_uuid) "
            "VALUES (?, ?, ?, ?, ?, ?)",
            (key, value, now, step, is_nan, run_id))
        conn.commit()
    finally:
        conn.close()


def log_metric_to_run(run_id: str, key: str, value: float, step: int = 0):
    """Log a metric onto an arbitrary run (used by aggregation)."""
    value = float(value)
    now = _now_ms()
    conn = _connect()
    try:
        conn.execute(
            "INSERT OR REPLACE INTO metrics "
            "(key, value, timestamp, run_uuid, step, is_nan) "
            "VALUES (?, ?, ?, ?, ?, ?)",
            (key, value, now, run_id, step, int(value != value)))
        conn.commit()
    finally:
        conn.close()


def log_image(image, key: str = "image", step: int = 0):
    """Save a PIL image under the run's artifact dir."""
    run_id = active_run_id()
    if run_id is None:
        raise RuntimeError("No active run")
    art_dir = os.path.join("mlruns", _EXPERIMENT["name"] if _EXPERIMENT
                           else "Default", run_id, "artifacts")
    os.makedirs(art_dir, exist_ok=True)
    image.save(os.path.join(art_dir, f"{key}_{step}.png"))


# ---------------------------------------------------------------------------
# Query API
# ---------------------------------------------------------------------------

def _parse_filter(filter_string: str):
    """Parse a conjunction of `tags.X = 'v'` / `params.X = 'v'` /
    `attributes.X = 'v'` clauses."""
    clauses = []
    if not filter_string:
        return clauses
    for part in filter_string.split(" and "):
        part = part.strip()
        if not part:
            continue
        lhs, rhs = part.split("=", 1)
        lhs = lhs.strip()
        rhs = rhs.strip().strip("'\"")
        kind, _, key = lhs.partition(".")
        clauses.append((kind, key, rhs))
    return clauses


def search_runs(experiment_names=None, experiment_ids=None,
                filter_string: str = "", max_results: int = 1000,
                output_format: str = "pandas"):
    """Return runs as a pandas DataFrame (mlflow.search_runs subset).

    Columns: run_id, experiment_id, status, start_time, end_time, plus
    params.<k>, tags.<k>, metrics.<k> (latest value) for all present keys.
    """
    import pandas as pd
    conn = _connect()
    try:
        _ensure_schema(conn)
        q = ("SELECT r.run_uuid, r.experiment_id, r.status, r.start_time, "
             "r.end_time FROM runs r JOIN experiments e "
             "ON r.experiment_id = e.experiment_id "
             "WHERE r.lifecycle_stage = 'active'")
        args = []
        if experiment_names:
            q += (" AND e.name IN (%s)" %
                  ",".join("?" * len(experiment_names)))
            args += list(experiment_names)
        if experiment_ids:
            q += (" AND r.experiment_id IN (%s)" %
                  ",".join("?" * len(experiment_ids)))
            args += [int(i) for i in experiment_ids]
        rows = conn.execute(q, args).fetchall()
        records = []
        for run_uuid, exp_id, status, st, et in rows:
            rec = {"run_id": run_uuid, "experiment_id": str(exp_id),
                   "status": status, "start_time": st, "end_time": et}
            for k, v in conn.execute(
                    "SELECT key, value FROM params WHERE run_uuid=?",
                    (run_uuid,)):
                rec[f"params.{k}"] = v
            for k, v in conn.execute(
                    "SELECT key, value FROM tags WHERE run_uuid=?",
                    (run_uuid,)):
                rec[f"tags.{k}"] = v
            for k, v in conn.execute(
                    "SELECT key, value FROM latest_metrics WHERE run_uuid=?",
                    (run_uuid,)):
                rec[f"metrics.{k}"] = v
            records.append(rec)
    finally:
        conn.close()

    for kind, key, val in _parse_filter(filter_string):
        col = {"tags": f"tags.{key}", "params": f"params.{key}",
               "attributes": key, "attribute": key}[kind]
        records = [r for r in records if str(r.get(col)) == val]

    records.sort(key=lambda r: r.get("start_time") or 0, reverse=True)
    records = records[:max_results]
    return pd.DataFrame(records)


def get_metric_history(run_id: str, key: str):
    """[(step, value)] for a metric, step-ascending."""
    conn = _connect()
    try:
        return conn.execute(
            "SELECT step, value FROM metrics WHERE run_uuid=? AND key=? "
            "ORDER BY step", (run_id, key)).fetchall()
    finally:
        conn.close()


def list_experiments():
    conn = _connect()
    try:
        _ensure_schema(conn)
        return conn.execute(
            "SELECT experiment_id, name FROM experiments "
            "WHERE lifecycle_stage='active'").fetchall()
    finally:
        conn.close()


def delete_run(run_id: str, hard: bool = False):
    conn = _connect()
    try:
        if hard:
            for t in ("metrics", "latest_metrics", "params", "tags"):
                conn.execute(f"DELETE FROM {t} WHERE run_uuid=?", (run_id,))
            conn.execute("DELETE FROM runs WHERE run_uuid=?", (run_id,))
        else:
            conn.execute(
                "UPDATE runs SET lifecycle_stage='deleted', deleted_time=? "
                "WHERE run_uuid=?", (_now_ms(), run_id))
        conn.commit()
    finally:
        conn.close()


def delete_experiment(experiment_id: int, hard: bool = False):
    conn = _connect()
    try:
        runs = [r[0] for r in conn.execute(
            "SELECT run_uuid FROM runs WHERE experiment_id=?",
            (experiment_id,)).fetchall()]
    finally:
        conn.close()
    for r in runs:
        delete_run(r, hard=hard)
    conn = _connect()
    try:
        if hard:
            conn.execute("DELETE FROM experiments WHERE experiment_id=?",
                         (experiment_id,))
        else:
            conn.execute(
                "UPDATE experiments SET lifecycle_stage='deleted' "
                "WHERE experiment_id=?", (experiment_id,))
        conn.commit()
    finally:
        conn.close()


@contextlib.contextmanager
def no_tracking():
    """Context manager that swallows logging (for --no-mlflow paths)."""
    yield
