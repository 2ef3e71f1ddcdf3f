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
