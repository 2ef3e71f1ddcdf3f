"""ActiveTesting with LURE risk estimates (Kossen et al. 2021).

Reference parity: coda/baselines/activetesting.py:9-122. Surrogate is the
ensemble mean; acquisition is expected loss summed over models, sampled
stochastically; risk is the LURE-weighted unbiased estimator.

The per-point surrogate loss sum over models (`1 - pi_y[n, argmax_h]`
summed over H) never changes, so it is precomputed once at init; the
reference rebuilds it from the full (H,N,C) tensor each step
(activetesting.py:31-44).
"""
from __future__ import annotations

import random

import torch

from .iid import IID
from .. import ops


class ActiveTesting(IID):
    def __init__(self, dataset, loss_fn):
        super().__init__(dataset, loss_fn)
        self.M = 0
        self.losses = []   # per-step (Hl,) loss rows
        self.qs = []       # sampling probabilities
        self.stochastic = True

        # surrogate = ensemble mean; static per-point acquisition mass:
        # sum_h (1 - pi_y[n, argmax_h(n)])
        pi_y = dataset.preds.mean(dim=0)                      # (N, C)
        y_star = pi_y.gather(1, self.classes.t())             # (N, Hl)
        self._acq_mass = (1.0 - y_star).sum(dim=1)            # (N,)

    def get_next_item_to_label(self):
        scores = self._acq_mass[self.d_u_idxs]
        scores = scores / scores.sum()
        # Draw with the seeded `random` module (reference parity:
        # activetesting.py:46) so trajectories are device-independent.
        weights = scores.cpu().tolist()
        local = random.choices(range(len(self.d_u_idxs)), weights=weights)[0]
        chosen_idx = self.d_u_idxs[local]
        return chosen_idx, weights[local]

    def get_vs(self) -> torch.Tensor:
        """LURE weights over the M sampled points (activetesting.py:52-67)."""
        qs = torch.tensor(self.qs, device=self.device, dtype=torch.float32)
        return ops.lure_weights(qs, self.N)

    def get_lure_risks_and_vars(self):
        losses = torch.stack(self.losses, dim=1).view(self.Hl, -1)  # (Hl, M)
        vs = self.get_vs().unsqueeze(0)                             # (1, M)
        weighted = vs * losses
        lure = weighted.mean(dim=1)
        var = weighted.var(dim=1, unbiased=True) / self.M
        return lure, var

    def add_label(self, chosen_idx, true_class, selection_prob=None):
        super().add_label(chosen_idx, true_class, selection_prob)
        self.losses.append(self._point_loss(int(chosen_idx), true_class))
        self.qs.append(selection_prob)
        self.M += 1

    def get_risk_estimates(self):
        lure, _ = self.get_lure_risks_and_vars()
        return lure

    def get_best_model_prediction(self):
        if self.losses:
            risk = self.get_risk_estimates()
            best_model_risk, best_model_idx_pred = torch.min(risk, dim=0)
            ties = risk == best_model_risk
            if int(ties.sum()) > 1:
                idxs = torch.nonzero(ties, as_tuple=True)[0]
                best_model_idx_pred = idxs[torch.randperm(len(idxs))[0]]
            return best_model_idx_pred
        return random.choice(list(range(self.Hl)))
