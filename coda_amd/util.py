"""Numeric sanitizers, ensemble consensus, debug plotting.

Behavioral counterpart of the reference's runtime guards (coda/util.py,
used throughout coda/coda.py's pbest pipeline): raise on non-finite
intermediates, validate probability rows, render bar charts for the
tracking store. Implemented independently; the guard granularity and
messages are this package's own.

The guards are opt-in via CODA_AMD_DEBUG=1 or `main.py --debug-checks`
(the reference ships them always-on; every check here costs a device
synchronization, ~1 ms/step at the headline config, so production runs
keep them off). The test suites enable them.
"""
from __future__ import annotations

import os

import torch

DEBUG = os.environ.get("CODA_AMD_DEBUG") == "1"


def set_debug(enabled: bool) -> None:
    """Flip the numeric guards at runtime (used by --debug-checks).

    The consuming modules bind DEBUG by value at import, so this updates
    each of them rather than only this module's flag.
    """
    global DEBUG
    DEBUG = enabled
    import coda_amd.ops as _ops
    import coda_amd.selectors.coda as _coda
    _ops.DEBUG = enabled
    _coda.DEBUG = enabled


class Ensemble:
    """Mean-ensemble consensus over the model axis (reference
    coda/util.py:7-14)."""

    def __init__(self, preds: torch.Tensor, **kwargs):
        self.preds = preds
        self.device = preds.device

    def get_preds(self, **kwargs) -> torch.Tensor:
        return self.preds.mean(dim=0)


def _check(t: torch.Tensor, name: str, *, raise_err: bool = True):
    """Guard a pipeline stage against NaN/Inf contamination.

    One fused finiteness reduction; diagnostics (count, value range) are
    only materialized on the failing path so the passing path costs a
    single device sync.
    """
    finite = torch.isfinite(t)
    if bool(finite.all()):
        return
    n_bad = int(t.numel() - finite.sum())
    lo = float(t.min())
    hi = float(t.max())
    msg = (f"non-finite tensor in stage '{name}': {n_bad}/{t.numel()} "
           f"elements, range [{lo:.3g}, {hi:.3g}]")
    if raise_err:
        raise RuntimeError(msg)
    print("[coda_amd check]", msg)


def _check_prob(p: torch.Tensor, name: str = "prob", eps: float = 1e-12):
    """Guard a tensor meant to hold probability rows (last axis sums to 1).

    Negative entries (beyond -eps) and non-finite row sums raise;
    drifted normalization only warns, matching the reference's tolerance
    for accumulated fp32 rounding in the Beta-integral rows.
    """
    _check(p, name)
    if bool((p < -eps).any()):
        raise RuntimeError(f"probability tensor '{name}' has entries < 0")
    row = p.sum(dim=-1)
    if not bool(torch.isfinite(row).all()):
        raise RuntimeError(f"probability rows of '{name}' sum to NaN/Inf")
    drift = (row - 1.0).abs().max()
    if float(drift) > 1e-4:
        print(f"[coda_amd check] '{name}' rows off-normalized by up to "
              f"{float(drift):.2e} (sums in [{float(row.min()):.4f}, "
              f"{float(row.max()):.4f}])")


def plot_bar(data, fig_size=(10, 5), title="", xlabel="", ylabel=""):
    """Render a 1-D series as a bar chart and return a PIL image.

    Used by the DEBUG_VIZ paths to log per-step EIG / P(best) charts
    into the tracking store (reference behavior: coda/coda.py:299-303).
    """
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    import numpy as np
    from PIL import Image

    if isinstance(data, torch.Tensor):
        values = data.detach().cpu().numpy()
    else:
        values = np.asarray(data)
    values = values.reshape(-1)

    fig, ax = plt.subplots(figsize=fig_size)
    try:
        ax.bar(np.arange(values.shape[0]), values)
        ax.set(title=title, xlabel=xlabel, ylabel=ylabel)
        fig.tight_layout()
        fig.canvas.draw()
        img = Image.frombuffer(
            "RGBA", fig.canvas.get_width_height(),
            fig.canvas.buffer_rgba()).convert("RGB")
    finally:
        plt.close(fig)
    return img
