"""Random (IID) acquisition baseline (reference: coda/baselines/iid.py).

Uniform random acquisition; risk = mean loss on the labeled set; random
tie-breaking for the best model. Also the base class for Uncertainty,
ActiveTesting and VMA.

Unlike the reference (which recomputes the loss over all labeled points
each call, iid.py:30-44), the per-model loss-sum is maintained
incrementally.
"""
from __future__ import annotations

import random

import torch

from ..base import ModelSelector
from .. import ops


class IID(ModelSelector):
    def __init__(self, dataset, loss_fn):
        self.Hl, self.N, self.C = dataset.preds.shape
        self.H = getattr(dataset, "total_models", self.Hl)
        self.d_l_idxs = []
        self.d_l_ys = []
        self.d_u_idxs = list(range(self.N))
        self.dataset = dataset
        self.device = dataset.preds.device
        self.loss_fn = loss_fn
        self.classes = ops.pred_classes(dataset.preds)  # cached (Hl, N)
        self._loss_sum = torch.zeros(self.Hl, device=self.device)
        self.stochastic = True

    def get_next_item_to_label(self):
        self.stochastic = True
        idx = random.choice(self.d_u_idxs)
        return idx, 1.0 / len(self.d_u_idxs)

    def _point_loss(self, idx: int, label: int) -> torch.Tensor:
        """(Hl,) loss of each model on one labeled point.

        Routed through self.loss_fn so a future non-accuracy loss stays
        consistent with the oracle; the accuracy loss keeps its
        cached-argmax fast path (no (Hl, C) gather from the pool).
        """
        from ..options import accuracy_loss
        if self.loss_fn is accuracy_loss or self.loss_fn is None:
            return 1.0 - (self.classes[:, idx] == label).float()
        point_preds = self.dataset.preds[:, idx].float()      # (Hl, C)
        label_t = torch.full((self.Hl,), int(label),
                             device=self.device, dtype=torch.long)
        return self.loss_fn(point_preds, label_t)

    def add_label(self, chosen_idx, true_class, selection_prob=None):
        chosen_idx = int(chosen_idx)
        self.d_u_idxs.remove(chosen_idx)
        self.d_l_idxs.append(chosen_idx)
        self.d_l_ys.append(true_class)
        self._loss_sum += self._point_loss(chosen_idx, true_class)

    def get_risk_estimates(self) -> torch.Tensor:
        """(Hl,) mean loss of each model over labeled points so far."""
        if not self.d_l_idxs:
            return torch.zeros(self.Hl, device=self.device)
        return self._loss_sum / len(self.d_l_idxs)

    def get_best_model_prediction(self):
        risk = self.get_risk_estimates()
        best_model_risk, best_model_idx_pred = torch.min(risk, dim=0)
        ties = risk == best_model_risk
        if int(ties.sum()) > 1:
            idxs = torch.nonzero(ties, as_tuple=True)[0]
            best_model_idx_pred = idxs[torch.randperm(len(idxs))[0]]
            self.stochastic = True
        return best_model_idx_pred
