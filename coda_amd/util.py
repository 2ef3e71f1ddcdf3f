"""Numeric sanitizers, ensemble consensus, debug plotting.

Reference parity: coda/util.py:7-66. The `_check`/`_check_prob` runtime
asserts are the de-facto correctness oracle for the math pipeline; they
are opt-in via CODA_AMD_DEBUG=1 (each check synchronizes the device; the
test suites turn them on).
"""
from __future__ import annotations

import os

import torch

# The reference ships its _DEBUG flag ON (coda/coda.py:10); here the
# equivalent guards are OPT-IN (CODA_AMD_DEBUG=1): every NaN/Inf check is
# a host-device synchronization (~5 per acquisition step measured, ~1 ms
# of the step at the headline config). The test suites enable them.
DEBUG = os.environ.get("CODA_AMD_DEBUG") == "1"


class Ensemble:
    """Mean-ensemble consensus over the model axis (coda/util.py:7-14)."""

    def __init__(self, preds: torch.Tensor, **kwargs):
        self.preds = preds
        self.device = preds.device

    def get_preds(self, **kwargs) -> torch.Tensor:
        return self.preds.mean(dim=0)


def _check(t: torch.Tensor, name: str, *, raise_err: bool = True):
    """Raise on NaN/Inf with min/max diagnostics."""
    bad = ~torch.isfinite(t)
    if bad.any():
        msg = (f"[NUMERIC ERROR] {name} has {int(bad.sum())} bad values "
               f"(NaN/Inf) out of {t.numel()} "
               f"min={t.min().item():.3g}, max={t.max().item():.3g}")
        if raise_err:
            raise RuntimeError(msg)
        print(msg)


def _check_prob(p: torch.Tensor, name: str = "prob", eps: float = 1e-12):
    """Validate p as a probability distribution along the last axis."""
    _check(p, name)
    if (p < -eps).any():
        raise RuntimeError(f"{name} has negatives")
    s = p.sum(-1)
    if (torch.isnan(s) | torch.isinf(s)).any():
        raise RuntimeError(f"{name} sum is nan/inf")
    if ((s - 1).abs() > 1e-4).any():
        print(f"[WARN] {name} rows not normalised: min sum={s.min():.4f}, "
              f"max sum={s.max():.4f}")


def plot_bar(data, fig_size=(10, 5), title="", xlabel="", ylabel=""):
    """Bar plot -> PIL image (for tracking-store image logging)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    import numpy as np
    from PIL import Image

    if isinstance(data, torch.Tensor):
        data = data.detach().cpu().numpy()
    data = np.asarray(data).squeeze()
    fig, ax = plt.subplots(figsize=fig_size)
    ax.bar(list(range(data.shape[0])), data)
    ax.set_title(title)
    ax.set_xlabel(xlabel)
    ax.set_ylabel(ylabel)
    plt.tight_layout()
    fig.canvas.draw()
    img = Image.frombuffer("RGBA", fig.canvas.get_width_height(),
                           fig.canvas.buffer_rgba()).convert("RGB")
    plt.close(fig)
    return img
