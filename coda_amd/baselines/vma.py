"""VMA: Variance Minimization for Active Model Selection (Matsuura & Hara 2023).

Reference parity: coda/baselines/vma.py:6-63 - acquisition proportional to
sum_{h'>h} |loss_h(x) - loss_h'(x)|, drawn stochastically; risk estimation
inherits LURE from ActiveTesting.

The reference materializes the (H, H, N_u) pairwise-difference tensor
(vma.py:31-41), an O(H^2 N) cliff at large H. This implementation uses the
order-statistics identity sum_{i<j} |x_i - x_j| = sum_k (2k - H + 1) x_(k)
(sorted), which is O(H log H) per point and never forms the pair tensor -
the same scores to fp32 rounding.
"""
from __future__ import annotations

import random

from .activetesting import ActiveTesting
from .. import ops


class VMA(ActiveTesting):
    def __init__(self, dataset, loss_fn):
        super().__init__(dataset, loss_fn)
        # Static per-point pairwise-difference mass (the surrogate and the
        # argmax classes never change).
        pi_y = dataset.preds.mean(dim=0)                  # (N, C)
        y_star = pi_y.gather(1, self.classes.t()).t()     # (Hl, N)
        self._vma_mass = ops.vma_pairwise(1.0 - y_star)   # (N,)

    def get_next_item_to_label(self):
        scores = self._vma_mass[self.d_u_idxs]
        total = scores.sum()
        if float(total) < 1e-12:
            chosen_idx = random.choice(self.d_u_idxs)
            return chosen_idx, 1.0 / len(self.d_u_idxs)
        scores = scores / total
        local = random.choices(range(len(self.d_u_idxs)),
                               weights=scores.cpu().tolist())[0]
        return self.d_u_idxs[local], float(scores[local])
