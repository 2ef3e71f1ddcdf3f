"""Baseline selectors: behavior and convergence."""
import random

import torch

from coda_amd import Oracle
from coda_amd.baselines import (IID, ActiveTesting, VMA, ModelPicker,
                                Uncertainty)
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.options import LOSS_FNS


def _easy_dataset():
    preds, labels = make_synthetic_task(H=6, N=400, C=4, seed=10,
                                        best_acc=0.95, worst_acc=0.4)
    return Dataset.from_tensors(preds, labels, "cpu")


def _run(selector, oracle, n):
    best = None
    for _ in range(n):
        idx, q = selector.get_next_item_to_label()
        selector.add_label(idx, oracle(idx), q)
        best = int(selector.get_best_model_prediction())
    return best


def test_iid_runs_and_converges():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    tl = oracle.true_losses(ds.preds)
    random.seed(0); torch.manual_seed(0)
    sel = IID(ds, LOSS_FNS["acc"])
    assert sel.stochastic
    best = _run(sel, oracle, 60)
    assert float(tl[best] - tl.min()) < 0.15


def test_iid_incremental_risk_matches_batch():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(1)
    sel = IID(ds, LOSS_FNS["acc"])
    for _ in range(10):
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(idx), q)
    # recompute from scratch (the reference's formulation, iid.py:30-44)
    risk = torch.zeros(sel.Hl)
    for idx, y in zip(sel.d_l_idxs, sel.d_l_ys):
        risk += LOSS_FNS["acc"](ds.preds[:, idx, :],
                                torch.tensor([y]).expand(sel.Hl))
    risk /= len(sel.d_l_idxs)
    torch.testing.assert_close(sel.get_risk_estimates(), risk)


def test_uncertainty_deterministic_and_picks_max_entropy():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    sel = Uncertainty(ds, LOSS_FNS["acc"])
    assert not sel.stochastic
    ent = -(ds.preds.mean(0) *
            torch.log(ds.preds.mean(0) + 1e-8)).sum(-1)
    idx, q = sel.get_next_item_to_label()
    assert abs(q - float(ent.max())) < 1e-6
    assert int(idx) == int(ent.argmax())


def test_activetesting_lure_unbiasedness_shape():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(2); torch.manual_seed(2)
    sel = ActiveTesting(ds, LOSS_FNS["acc"])
    for _ in range(15):
        idx, q = sel.get_next_item_to_label()
        assert 0 < q <= 1
        sel.add_label(idx, oracle(idx), q)
    lure, var = sel.get_lure_risks_and_vars()
    assert lure.shape == (sel.Hl,) and var.shape == (sel.Hl,)
    assert torch.isfinite(lure).all() and torch.isfinite(var).all()
    tl = oracle.true_losses(ds.preds)
    best = _run(sel, oracle, 40)
    assert float(tl[best] - tl.min()) < 0.2


def test_vma_matches_bruteforce_acquisition():
    ds = _easy_dataset()
    random.seed(3)
    sel = VMA(ds, LOSS_FNS["acc"])
    # brute-force the reference's O(H^2) masses (vma.py:31-41)
    pi_y = ds.preds.mean(0)
    cls = ds.preds.argmax(-1)
    y_star = pi_y.gather(1, cls.t()).t()
    losses = 1 - y_star
    diff = (losses.unsqueeze(0) - losses.unsqueeze(1)).abs()
    mask = torch.triu(torch.ones(sel.Hl, sel.Hl, dtype=torch.bool), 1)
    want = diff[mask].sum(0)
    torch.testing.assert_close(sel._vma_mass, want, rtol=1e-4, atol=1e-5)


def test_vma_runs():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(4); torch.manual_seed(4)
    sel = VMA(ds, LOSS_FNS["acc"])
    _run(sel, oracle, 10)
    assert sel.M == 10


def test_modelpicker_posterior_and_convergence():
    ds = _easy_dataset()
    oracle = Oracle(ds, LOSS_FNS["acc"])
    tl = oracle.true_losses(ds.preds)
    random.seed(5); torch.manual_seed(5)
    sel = ModelPicker(ds, epsilon=0.45)
    for _ in range(40):
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(idx), q)
        assert abs(float(sel.posterior.sum()) - 1.0) < 1e-4
    best = int(sel.get_best_model_prediction())
    assert float(tl[best] - tl.min()) < 0.15


def test_modelpicker_entropies_match_classloop():
    """Batched entropy == the reference's per-class loop
    (modelpicker.py:74-86)."""
    ds = _easy_dataset()
    sel = ModelPicker(ds, epsilon=0.45)
    preds_u = sel.classes_nh[:50]
    got = sel.compute_entropies(preds_u, sel.posterior, sel.Hl, sel.C,
                                sel.gamma)
    # class loop
    want = torch.zeros(50)
    post = sel.posterior.unsqueeze(0).expand(50, sel.Hl)
    for c in range(sel.C):
        agree = (preds_u == c).float()
        np_ = post * (sel.gamma ** agree)
        np_ = np_ / np_.sum(1, keepdim=True)
        p = np_.clamp(min=1e-12)
        want += -(p * torch.log2(p)).sum(1) / sel.C
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
