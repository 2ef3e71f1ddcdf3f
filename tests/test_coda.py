"""CODA selector behavior: convergence, determinism, ablations, prefilter."""
import random

import pytest
import torch

from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.options import LOSS_FNS


def _run_steps(selector, oracle, n):
    choices = []
    for _ in range(n):
        idx, q = selector.get_next_item_to_label()
        selector.add_label(idx, oracle(idx), q)
        choices.append((int(idx), float(q),
                        int(selector.get_best_model_prediction())))
    return choices


def test_converges_to_planted_best():
    """On an easy task with a clearly-best model, CODA finds it quickly."""
    preds, labels = make_synthetic_task(H=6, N=400, C=4, seed=1,
                                        best_acc=0.95, worst_acc=0.45)
    ds = Dataset.from_tensors(preds, labels, "cpu")
    oracle = Oracle(ds, LOSS_FNS["acc"])
    true_losses = oracle.true_losses(ds.preds)
    best = int(true_losses.argmin())

    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, chunk_size=128)
    choices = _run_steps(sel, oracle, 15)
    # converged: the last 5 predictions are the true best model
    assert all(c[2] == best for c in choices[-5:]), (choices, best)


def test_deterministic_given_seed(synthetic_dataset):
    ds = synthetic_dataset
    oracle = Oracle(ds, LOSS_FNS["acc"])

    def traj():
        random.seed(7); torch.manual_seed(7)
        sel = CODA(ds, chunk_size=64)
        return _run_steps(sel, oracle, 5)

    t1, t2 = traj(), traj()
    assert t1 == t2


def test_pbest_is_distribution(synthetic_dataset):
    sel = CODA(synthetic_dataset)
    p = sel.get_pbest()
    assert p.shape == (synthetic_dataset.preds.shape[0],)
    assert torch.isfinite(p).all()
    assert abs(float(p.sum()) - 1.0) < 1e-4


def test_add_label_updates_posterior(synthetic_dataset):
    sel = CODA(synthetic_dataset)
    d0 = sel.dirichlets.clone()
    sel.add_label(3, 2, 0.5)
    assert not torch.equal(d0, sel.dirichlets)
    # only the true-class row moved, by update_strength at argmax positions
    diff = sel.dirichlets - d0
    assert diff[:, [0, 1, 3, 4]].abs().max() == 0
    assert torch.allclose(diff[:, 2].sum(-1),
                          torch.full((sel.Hl,), sel.update_strength))
    assert 3 in sel.labeled_idxs and 3 not in sel.unlabeled_idxs


def test_prefilter_drops_unanimous_points():
    preds, labels = make_synthetic_task(H=4, N=100, C=3, seed=2)
    # make points 0..9 unanimous: every model predicts class 0 strongly
    preds[:, :10] = 0.0
    preds[:, :10, 0] = 1.0
    ds = Dataset.from_tensors(preds, labels, "cpu")
    sel = CODA(ds)
    kept = sel._prefilter(list(range(100)))
    assert all(i >= 10 for i in kept)


def test_prefilter_n_subsamples_and_sets_stochastic(synthetic_dataset):
    random.seed(0)
    sel = CODA(synthetic_dataset, prefilter_n=20)
    kept = sel._prefilter(list(range(synthetic_dataset.preds.shape[1])))
    assert len(kept) == 20
    assert sel.stochastic


@pytest.mark.parametrize("q", ["iid", "uncertainty"])
def test_q_ablations_run(synthetic_dataset, q):
    random.seed(0); torch.manual_seed(0)
    sel = CODA(synthetic_dataset, q=q)
    oracle = Oracle(synthetic_dataset, LOSS_FNS["acc"])
    choices = _run_steps(sel, oracle, 3)
    assert len(choices) == 3


def test_no_diag_prior_ablation(synthetic_dataset):
    sel = CODA(synthetic_dataset, disable_diag_prior=True)
    p = sel.get_pbest()
    assert torch.isfinite(p).all()


def test_chunk_size_invariance(synthetic_dataset):
    """EIG values must not depend on the candidate chunk size."""
    oracle = Oracle(synthetic_dataset, LOSS_FNS["acc"])
    random.seed(3); torch.manual_seed(3)
    s1 = CODA(synthetic_dataset, chunk_size=17)
    e1, c1 = s1.eig_batched()
    random.seed(3); torch.manual_seed(3)
    s2 = CODA(synthetic_dataset, chunk_size=300)
    e2, c2 = s2.eig_batched()
    assert c1 == c2
    torch.testing.assert_close(e1, e2, rtol=1e-6, atol=1e-7)


@pytest.mark.parametrize("dtype", ["bf16", "fp8"])
def test_low_precision_storage(synthetic_task, dtype, tmp_path):
    """CODA runs on bf16/fp8 storage; trajectories match fp32 on a
    well-separated task (compute always up-casts to fp32)."""
    from coda_amd.datasets import write_synthetic_task
    path = write_synthetic_task(str(tmp_path), name="lp", H=6, N=200, C=4,
                                seed=9, best_acc=0.95, worst_acc=0.4)

    def run(storage):
        ds = Dataset(path, "cpu", storage_dtype=storage)
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=64)
        traj = []
        for _ in range(3):
            idx, q = sel.get_next_item_to_label()
            sel.add_label(idx, oracle(int(idx)), q)
            traj.append(int(idx))
        return traj, int(sel.get_best_model_prediction())

    t32, b32 = run("fp32")
    tlp, blp = run(dtype)
    assert blp == b32
    # bf16 rounds probabilities to ~3 decimal digits; selections can
    if dtype == "bf16":
        assert tlp == t32


def test_incremental_pi_hat_matches_full():
    """The rank-1 incremental pi_hat maintained by add_label equals a full
    recomputation from the updated Dirichlets (to fp32 roundoff)."""
    from coda_amd import ops
    preds, labels = make_synthetic_task(H=6, N=250, C=5, seed=13)
    ds = Dataset.from_tensors(preds, labels, "cpu")
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, chunk_size=64)
    for _ in range(5):
        i, q = sel.get_next_item_to_label()
        sel.add_label(i, oracle(int(i)), q)
    full = ops.pi_hat_partial(sel.dirichlets, ds.preds)
    torch.testing.assert_close(sel._adjusted, full, rtol=1e-4, atol=1e-5)
    xi_full, pi_full = ops.pi_hat_normalize(full)
    torch.testing.assert_close(sel.pi_hat, pi_full, rtol=1e-4, atol=1e-6)


class TestEdgeShapes:
    """Degenerate and boundary shapes the engine must survive."""

    @pytest.mark.parametrize("H,C", [(2, 2), (2, 5), (3, 2), (9, 3)])
    def test_small_pools_and_binary_tasks(self, H, C):
        preds, labels = make_synthetic_task(H=H, N=80, C=C, seed=40)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=32)
        for _ in range(3):
            idx, q = sel.get_next_item_to_label()
            sel.add_label(idx, oracle(int(idx)), q)
        p = sel.get_pbest()
        assert torch.isfinite(p).all() and p.shape == (H,)

    def test_all_models_agree_everywhere(self):
        """Unanimous pool: the prefilter drops everything and the
        fallback-to-all-unlabeled path (coda/coda.py:239 `or`) engages."""
        preds = torch.zeros(4, 50, 3)
        preds[:, :, 1] = 1.0
        labels = torch.ones(50, dtype=torch.long)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=16)
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(int(idx)), q)
        assert torch.isfinite(sel.get_pbest()).all()

    def test_nearly_exhausted_pool(self):
        preds, labels = make_synthetic_task(H=4, N=12, C=3, seed=41)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=8)
        for _ in range(10):
            idx, q = sel.get_next_item_to_label()
            sel.add_label(idx, oracle(int(idx)), q)
        assert len(sel.unlabeled_idxs) == 2

    def test_nonstandard_num_points(self):
        """num_points != 256 runs the eager grid (GPU included)."""
        preds, labels = make_synthetic_task(H=4, N=60, C=3, seed=42)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        sel = CODA(ds, chunk_size=16, num_points=128, eig_impl="fused")
        e, c = sel.eig_batched()
        assert torch.isfinite(e).all()
