"""Ground-truth label oracle (reference: coda/oracle.py:1-24).

Serves true labels point-by-point and computes each model's true mean loss
(used only for regret evaluation). The per-model loss uses the cached argmax
classes instead of re-deriving them from the full (H, N, C) tensor each call.
"""
from __future__ import annotations

import torch

from . import ops


class Oracle:
    def __init__(self, dataset, loss_fn=None):
        self.dataset = dataset
        self.loss_fn = loss_fn
        self.device = dataset.device
        self.labels = dataset.labels
        assert self.labels is not None, "Oracle needs labels!"
        # host-side copy: label serving is a per-step scalar fetch and
        # must not synchronize the device (reference syncs every step,
        # oracle.py:24)
        self._labels_host = self.labels.cpu().tolist()

    def true_losses(self, preds: torch.Tensor) -> torch.Tensor:
        """Mean loss per model: (H, N, C) post-softmax scores -> (H,)."""
        if self.loss_fn is None or getattr(self.loss_fn, "__name__", "") == "accuracy_loss":
            return ops.accuracy_losses(ops.pred_classes(preds), self.labels)
        H, N, C = preds.shape
        return self.loss_fn(preds.reshape(-1, C), self.labels.repeat(H),
                            reduction="none").view(H, N).mean(dim=1)

    def __call__(self, idx) -> int:
        return int(self._labels_host[int(idx)])
