"""Max-ensemble-entropy acquisition (reference: coda/baselines/uncertainty.py).

Picks the unlabeled point whose ensemble-mean prediction has the highest
entropy. Non-adaptive: the per-point entropy never changes, so it is
computed ONCE at init (the reference recomputes the full (N,C) entropy
every step, uncertainty.py:6-11,38).
"""
from __future__ import annotations

import torch

from .iid import IID
from .. import ops


def uncertainty(preds: torch.Tensor, d_u_idxs) -> torch.Tensor:
    """Reference-parity helper: entropy of the ensemble mean on a subset."""
    ent = ops.entropy_acquisition(preds.mean(dim=0))
    return ent[d_u_idxs]


class Uncertainty(IID):
    def __init__(self, dataset, loss_fn):
        super().__init__(dataset, loss_fn)
        self.stochastic = False
        self._entropy = ops.entropy_acquisition(dataset.preds.mean(dim=0))

    def get_next_item_to_label(self):
        ent = self._entropy[self.d_u_idxs]
        chosen_q, chosen_idx_local = torch.max(ent, dim=0)
        ties = ent == chosen_q
        if int(ties.sum()) > 1:
            self.stochastic = True
            idxs = torch.nonzero(ties, as_tuple=True)[0]
            chosen_idx_local = idxs[torch.randperm(len(idxs))[0]]
        return self.d_u_idxs[chosen_idx_local], float(chosen_q)
