"""Full-harness e2e on the GPU: CLI -> HIP engine -> tracking DB.

Marked gpu; gives the round-end GPU suite coverage of the complete
user-facing loop (not just the kernels)."""
import sqlite3

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _require_gpu():
    assert torch.cuda.is_available()


def test_main_cli_end_to_end_gpu(tmp_path, monkeypatch):
    from coda_amd.datasets import write_synthetic_task
    from coda_amd import tracking
    import main as harness

    write_synthetic_task(str(tmp_path / "data"), name="gtask", H=8, N=500,
                         C=10, seed=3, best_acc=0.93, worst_acc=0.5)
    monkeypatch.chdir(tmp_path)
    tracking.set_tracking_uri("sqlite:///coda.sqlite")
    tracking._EXPERIMENT = None
    tracking._RUN_STACK.clear()

    harness.main(["--task", "gtask", "--data-dir", "data", "--method",
                  "coda", "--iters", "8", "--seeds", "1",
                  "--chunk-size", "128"])

    conn = sqlite3.connect(str(tmp_path / "coda.sqlite"))
    rows = conn.execute(
        "SELECT m.step, m.value FROM metrics m "
        "JOIN tags t ON m.run_uuid = t.run_uuid "
        "AND t.key='mlflow.parentRunId' "
        "WHERE m.key='regret' ORDER BY m.step").fetchall()
    times = conn.execute(
        "SELECT value FROM metrics WHERE key='step_seconds'").fetchall()
    conn.close()
    assert len(rows) == 8
    # converged on an easy planted-best task by the last steps
    assert rows[-1][1] < 0.05
    assert len(times) == 8 and all(t[0] > 0 for t in times)


def test_serve_session_gpu(tmp_path):
    """Interactive serving against the GPU engine."""
    from coda_amd.datasets import Dataset, make_synthetic_task
    from coda_amd.oracle import Oracle
    from coda_amd.options import LOSS_FNS
    from coda_amd.serve import SelectorSession

    preds, labels = make_synthetic_task(H=6, N=300, C=5, seed=4)
    ds = Dataset.from_tensors(preds, labels, "cuda:0")
    oracle = Oracle(ds, LOSS_FNS["acc"])
    sess = SelectorSession(ds, method="coda", oracle=oracle, chunk_size=64)
    for _ in range(3):
        idx, _ = sess.next_item()
        sess.answer(idx, oracle(idx))
    st = sess.state()
    assert st["step"] == 3
    pb = sess.pbest()
    assert abs(sum(pb) - 1.0) < 1e-3
