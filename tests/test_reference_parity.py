"""Trajectory parity against the upstream reference implementation.

The reference (justinkay/coda) is pure PyTorch, so when its source tree is
mounted (dev containers only) we can run BOTH engines on the same synthetic
task with the same seeds and require identical selection trajectories.
Skipped automatically where /root/reference is absent (CI / GPU boxes).
"""
import contextlib
import io
import os
import random
import sys

import pytest
import torch

REF = "/root/reference"
pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "coda")),
    reason="reference tree not mounted")


@pytest.fixture(scope="module")
def ref_modules():
    sys.path.insert(0, REF)
    try:
        import coda.coda as ref_coda  # noqa
        import coda.baselines.modelpicker as ref_mp  # noqa
        yield ref_coda, ref_mp
    finally:
        sys.path.remove(REF)


class _RefDS:
    pass


def _mk(seed=0, H=8, N=300, C=5):
    from coda_amd.datasets import make_synthetic_task
    return make_synthetic_task(H=H, N=N, C=C, seed=seed)


def test_coda_trajectory_parity(ref_modules):
    ref_coda, _ = ref_modules
    from coda_amd.datasets import Dataset
    from coda_amd import CODA

    preds, labels = _mk()
    rds = _RefDS()
    rds.preds, rds.labels, rds.device = preds.clone(), labels.clone(), \
        torch.device("cpu")

    random.seed(0); torch.manual_seed(0)
    ref = ref_coda.CODA(rds)
    random.seed(0); torch.manual_seed(0)
    mine = CODA(Dataset.from_tensors(preds, labels, "cpu"))

    torch.testing.assert_close(ref.get_pbest().reshape(-1),
                               mine.get_pbest().reshape(-1),
                               rtol=1e-4, atol=1e-6)

    for m in range(5):
        random.seed(100 + m)
        with contextlib.redirect_stderr(io.StringIO()):
            ir, qr = ref.get_next_item_to_label()
        random.seed(100 + m)
        im, qm = mine.get_next_item_to_label()
        assert int(ir) == int(im), (m, int(ir), int(im))
        assert abs(qr - qm) < 1e-5
        y = int(labels[int(ir)])
        ref.add_label(int(ir), y, qr)
        mine.add_label(int(im), y, qm)
        assert int(ref.get_best_model_prediction()) == \
            int(mine.get_best_model_prediction())


def test_modelpicker_entropy_parity(ref_modules):
    _, ref_mp = ref_modules
    from coda_amd.datasets import Dataset
    from coda_amd.baselines import ModelPicker

    preds, labels = _mk(seed=3)
    rds = _RefDS()
    rds.preds, rds.labels, rds.device = preds.clone(), labels.clone(), \
        torch.device("cpu")
    ref = ref_mp.ModelPicker(rds, epsilon=0.45)
    mine = ModelPicker(Dataset.from_tensors(preds, labels, "cpu"),
                       epsilon=0.45)

    preds_u = ref.dataset.preds.argmax(2).transpose(0, 1)[:60]
    e_ref = ref.compute_entropies(preds_u, ref.posterior, ref.H, ref.C,
                                  ref.gamma)
    e_mine = mine.compute_entropies(preds_u, mine.posterior, mine.Hl,
                                    mine.C, mine.gamma)
    torch.testing.assert_close(e_ref, e_mine, rtol=1e-5, atol=1e-6)


def test_oracle_losses_parity(ref_modules):
    from coda_amd.datasets import Dataset
    from coda_amd.oracle import Oracle
    from coda_amd.options import LOSS_FNS
    sys.path.insert(0, REF)
    try:
        from coda.oracle import Oracle as RefOracle
        from coda.options import LOSS_FNS as REF_LOSS
    finally:
        sys.path.remove(REF)

    preds, labels = _mk(seed=4)
    rds = _RefDS()
    rds.preds, rds.labels, rds.device = preds.clone(), labels.clone(), \
        torch.device("cpu")
    ref = RefOracle(rds, loss_fn=REF_LOSS["acc"])
    mine = Oracle(Dataset.from_tensors(preds, labels, "cpu"),
                  loss_fn=LOSS_FNS["acc"])
    torch.testing.assert_close(ref.true_losses(preds),
                               mine.true_losses(preds))


def test_coda_prefilter_and_ablation_parity(ref_modules):
    """Prefiltered candidates + no-diag-prior ablation match the
    reference's trajectories too."""
    ref_coda, _ = ref_modules
    from coda_amd.datasets import Dataset
    from coda_amd import CODA

    preds, labels = _mk(seed=6, H=6, N=250, C=4)
    rds = _RefDS()
    rds.preds, rds.labels, rds.device = preds.clone(), labels.clone(), \
        torch.device("cpu")

    random.seed(3); torch.manual_seed(3)
    ref = ref_coda.CODA(rds, prefilter_n=40, disable_diag_prior=True)
    random.seed(3); torch.manual_seed(3)
    mine = CODA(Dataset.from_tensors(preds, labels, "cpu"),
                prefilter_n=40, disable_diag_prior=True)

    for m in range(3):
        random.seed(50 + m)
        with contextlib.redirect_stderr(io.StringIO()):
            ir, qr = ref.get_next_item_to_label()
        random.seed(50 + m)
        im, qm = mine.get_next_item_to_label()
        assert int(ir) == int(im), (m, int(ir), int(im))
        y = int(labels[int(ir)])
        ref.add_label(int(ir), y, qr)
        mine.add_label(int(im), y, qm)


def test_baseline_acquisition_distributions_parity(ref_modules):
    """ActiveTesting / VMA acquisition DISTRIBUTIONS equal the
    reference's (selection draws consume the same seeded RNG, so equal
    distributions => equal trajectories)."""
    sys.path.insert(0, REF)
    try:
        from coda.baselines.activetesting import ActiveTesting as RefAT
        from coda.baselines.vma import VMA as RefVMA
        from coda.options import LOSS_FNS as REF_LOSS
    finally:
        sys.path.remove(REF)
    from coda_amd.datasets import Dataset
    from coda_amd.baselines import ActiveTesting, VMA
    from coda_amd.options import LOSS_FNS

    preds, labels = _mk(seed=7, H=5, N=120, C=4)
    rds = _RefDS()
    rds.preds, rds.labels, rds.device = preds.clone(), labels.clone(), \
        torch.device("cpu")
    ds = Dataset.from_tensors(preds, labels, "cpu")

    # ActiveTesting: same normalized scores over the unlabeled pool
    rat = RefAT(rds, REF_LOSS["acc"])
    mat = ActiveTesting(ds, LOSS_FNS["acc"])
    pi_y = rds.preds.mean(0)
    cls = rds.preds.argmax(2)
    y_star = pi_y[torch.arange(120), cls]
    ref_scores = (1 - y_star).sum(0)
    ref_scores = ref_scores / ref_scores.sum()
    mine_scores = mat._acq_mass / mat._acq_mass.sum()
    torch.testing.assert_close(mine_scores, ref_scores, rtol=1e-5,
                               atol=1e-7)
    # identical seeded draw
    random.seed(11)
    ri, rq = rat.get_next_item_to_label()
    random.seed(11)
    mi, mq = mat.get_next_item_to_label()
    assert int(ri) == int(mi) and abs(rq - mq) < 1e-6

    # VMA: same pairwise-difference masses (ours via the sorted identity)
    rv = RefVMA(rds, REF_LOSS["acc"])
    mv = VMA(ds, LOSS_FNS["acc"])
    random.seed(12)
    ri, rq = rv.get_next_item_to_label()
    random.seed(12)
    mi, mq = mv.get_next_item_to_label()
    assert int(ri) == int(mi) and abs(rq - mq) < 1e-6
