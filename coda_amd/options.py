"""Loss registry (reference: coda/options.py:1-18)."""
from __future__ import annotations

import torch


def accuracy_loss(preds, labels, **kwargs):
    """1 - accuracy, non-reduced. Accepts score matrices or integer labels."""
    if len(labels.shape) > 1:
        accs = (torch.argmax(preds, dim=-1) == torch.argmax(labels, dim=-1)).float()
    else:
        accs = (torch.argmax(preds, dim=-1) == labels).float()
    return 1.0 - accs


LOSS_FNS = {
    "acc": accuracy_loss,
    # 'ce' is unsupported in the reference too (no logits in the data format).
}
