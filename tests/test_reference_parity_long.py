"""Paper-protocol-length trajectory parity vs the mounted reference.

Extends tests/test_reference_parity.py from 5-step smoke parity to
100-step trajectories (the reference experiment length, main.py:35) on
three shapes, including the C=126 (DomainNet-like) and C=2 (binary
GLUE/camelyon-like) class-count edges of the benchmark suite
(reference paper/fig3.py:129-193 groups), and to all six methods.

Comparison protocol: at several of these shapes the EIG signal sits
within a few fp32 ulps of the H_before cancellation (measured: values
~5e-6 with 4.8e-7 quantization at C=126), so SELECTION equality is not
a well-posed target - any summation reorder flips seeded tie-breaks.
Instead each step asserts the full ACQUISITION SCORE VECTORS agree to
tight tolerance, then advances BOTH engines along the reference's own
choice (teacher forcing), so the dynamics stay comparable for the full
100 steps. Methods whose selection involves no fp comparison (IID) are
held to exact selection equality.

Runs only where /root/reference is mounted (dev container); the
reference engine is the slow side (~1 s/step at C=126 on CPU).
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
def refpkg():
    sys.path.insert(0, REF)
    try:
        import coda.coda as ref_coda
        import coda.baselines.iid as ref_iid
        import coda.baselines.uncertainty as ref_unc
        import coda.baselines.activetesting as ref_at
        import coda.baselines.vma as ref_vma
        import coda.baselines.modelpicker as ref_mp
        from coda.options import LOSS_FNS as REF_LOSS
        yield dict(coda=ref_coda, iid=ref_iid, unc=ref_unc, at=ref_at,
                   vma=ref_vma, mp=ref_mp, loss=REF_LOSS)
    finally:
        sys.path.remove(REF)


class _RefDS:
    pass


def _task(H, N, C, seed):
    from coda_amd.datasets import make_synthetic_task
    preds, labels = make_synthetic_task(H=H, N=N, C=C, seed=seed)
    rds = _RefDS()
    rds.preds, rds.labels = preds.clone(), labels.clone()
    rds.device = torch.device("cpu")
    from coda_amd.datasets import Dataset
    return rds, Dataset.from_tensors(preds, labels, "cpu"), labels


def _ref_select(q_vals, candidates, seed):
    """The reference's greedy selection rule (coda/coda.py:306-313)
    applied to a given EIG vector, with seeded tie randomization."""
    random.seed(seed)
    best = q_vals.max()
    ties = torch.isclose(q_vals, best, rtol=1e-8)
    if int(ties.sum()) > 1:
        pos = random.choice(torch.nonzero(ties, as_tuple=True)[0].tolist())
    else:
        pos = int(q_vals.argmax())
    return int(list(candidates)[pos]), float(q_vals[pos])


@pytest.mark.timeout(1200)
@pytest.mark.parametrize("H,N,C,steps", [
    (8, 300, 5, 100),     # mid class count (MSV/cifar-like)
    (6, 400, 2, 100),     # binary edge (glue/camelyon group)
    (10, 120, 126, 100),  # DomainNet-126 class-count edge
])
def test_coda_100step_trajectory_parity(refpkg, H, N, C, steps):
    from coda_amd import CODA

    rds, ds, labels = _task(H, N, C, seed=H + C)
    random.seed(0); torch.manual_seed(0)
    ref = refpkg["coda"].CODA(rds)
    random.seed(0); torch.manual_seed(0)
    mine = CODA(ds)

    for m in range(steps):
        with contextlib.redirect_stderr(io.StringIO()):
            q_r, cand_r = ref.eig_batched()
        q_m, cand_m = mine.eig_batched()
        assert list(cand_r) == list(cand_m), m
        # EIG parity: EIG = H_before - (candidate sum) is a
        # cancellation at H_before scale (~log2 H), so the reorder
        # noise floor is a few ulps OF H_BEFORE, not of the EIG value:
        # ~16 ulps = 4e-6 here. Long-horizon drift is caught by the
        # pbest check below (O(1/H) scale, rtol 1e-3).
        torch.testing.assert_close(q_m, q_r, rtol=1e-3, atol=4e-6)
        # advance both along the reference's own (seeded) selection
        idx, q = _ref_select(q_r, cand_r, seed=1000 + m)
        y = int(labels[idx])
        ref.add_label(idx, y, q)
        mine.add_label(idx, y, q)
        if (m + 1) % 10 == 0:
            torch.testing.assert_close(mine.get_pbest().reshape(-1),
                                       ref.get_pbest().reshape(-1),
                                       rtol=1e-3, atol=1e-6)


@pytest.mark.timeout(600)
def test_iid_100step_exact(refpkg):
    """IID selection is pure seeded RNG over the same list - exact."""
    from coda_amd.baselines import IID
    from coda_amd.options import LOSS_FNS

    rds, ds, labels = _task(H=7, N=250, C=6, seed=21)
    ref = refpkg["iid"].IID(rds, refpkg["loss"]["acc"])
    mine = IID(ds, LOSS_FNS["acc"])
    for m in range(100):
        random.seed(1000 + m)
        ir, qr = ref.get_next_item_to_label()
        random.seed(1000 + m)
        im, qm = mine.get_next_item_to_label()
        assert int(ir) == int(im) and abs(qr - qm) < 1e-9
        y = int(labels[int(ir)])
        ref.add_label(int(ir), y, qr)
        mine.add_label(int(im), y, qm)
        torch.testing.assert_close(mine.get_risk_estimates(),
                                   ref.get_risk_estimates())


@pytest.mark.timeout(900)
@pytest.mark.parametrize("method", ["uncertainty", "activetesting",
                                    "vma", "model_picker"])
def test_baseline_100step_trajectory_parity(refpkg, method):
    """AT/VMA: same seeded stochastic draws (identical acquisition
    distributions); risk estimates compared as vectors each step (their
    best-model argmin has exact ties early on, where either pick is
    correct). Uncertainty/ModelPicker: deterministic scores compared,
    reference's pick teacher-forced."""
    from coda_amd.baselines import (Uncertainty, ActiveTesting, VMA,
                                    ModelPicker)
    from coda_amd.options import LOSS_FNS

    rds, ds, labels = _task(H=7, N=250, C=6, seed=21)
    if method == "uncertainty":
        ref = refpkg["unc"].Uncertainty(rds, refpkg["loss"]["acc"])
        mine = Uncertainty(ds, LOSS_FNS["acc"])
    elif method == "activetesting":
        ref = refpkg["at"].ActiveTesting(rds, refpkg["loss"]["acc"])
        mine = ActiveTesting(ds, LOSS_FNS["acc"])
    elif method == "vma":
        ref = refpkg["vma"].VMA(rds, refpkg["loss"]["acc"])
        mine = VMA(ds, LOSS_FNS["acc"])
    else:
        ref = refpkg["mp"].ModelPicker(rds, epsilon=0.38)
        mine = ModelPicker(ds, epsilon=0.38)

    flips = 0
    for m in range(100):
        random.seed(1000 + m); torch.manual_seed(1000 + m)
        with contextlib.redirect_stderr(io.StringIO()):
            ir, qr = ref.get_next_item_to_label()
        random.seed(1000 + m); torch.manual_seed(1000 + m)
        im, qm = mine.get_next_item_to_label()
        if int(ir) != int(im):
            # deterministic argmin/argmax near-tie: both engines must
            # consider the two picks equivalent
            flips += 1
            assert abs(qr - qm) < 1e-4 * max(1.0, abs(qr)), (m, qr, qm)
        y = int(labels[int(ir)])
        ref.add_label(int(ir), y, qr)
        mine.add_label(int(ir), y, qr)      # teacher-force ref's pick
        if method == "model_picker":
            torch.testing.assert_close(mine.posterior, ref.posterior,
                                       rtol=1e-5, atol=1e-7)
        elif method in ("activetesting", "vma"):
            r_risk, _ = ref.get_lure_risks_and_vars()
            m_risk, _ = mine.get_lure_risks_and_vars()
            torch.testing.assert_close(m_risk, r_risk, rtol=1e-4,
                                       atol=1e-6)
    # near-tie flips are expected for argmin-style methods whose score
    # landscape has exact plateaus (ModelPicker entropy over 6 classes);
    # the per-step posterior/risk closeness above is the parity signal
    assert flips <= 25, f"{flips} near-tie selection flips in 100 steps"


@pytest.mark.timeout(600)
def test_coda_hyperparameter_grid_parity(refpkg):
    """Non-default hyperparameters (lr, alpha, multiplier) track the
    reference for 25 steps each."""
    from coda_amd import CODA

    for kwargs in ({"learning_rate": 0.1},
                   {"alpha": 0.5},
                   {"multiplier": 1.0},
                   {"learning_rate": 0.05, "alpha": 0.8,
                    "multiplier": 4.0}):
        rds, ds, labels = _task(H=6, N=200, C=4, seed=33)
        random.seed(0); torch.manual_seed(0)
        ref = refpkg["coda"].CODA(rds, **kwargs)
        random.seed(0); torch.manual_seed(0)
        mine = CODA(ds, **kwargs)
        for m in range(25):
            with contextlib.redirect_stderr(io.StringIO()):
                q_r, cand_r = ref.eig_batched()
            q_m, cand_m = mine.eig_batched()
            assert list(cand_r) == list(cand_m)
            torch.testing.assert_close(q_m, q_r, rtol=1e-3, atol=2e-6)
            idx, q = _ref_select(q_r, cand_r, seed=500 + m)
            y = int(labels[idx])
            ref.add_label(idx, y, q)
            mine.add_label(idx, y, q)
        torch.testing.assert_close(mine.get_pbest().reshape(-1),
                                   ref.get_pbest().reshape(-1),
                                   rtol=1e-3, atol=1e-6)
