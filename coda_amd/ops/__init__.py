"""Compute-op dispatch: CDNA4 HIP kernels on GPU, eager PyTorch on CPU.

The hot ops (the Beta-grid P(best) integral and the fused EIG pipeline,
>95% of wall time in an acquisition step - reference hot loop at
coda/coda.py:235-281) dispatch to hand-written gfx950 HIP kernels from
`coda_amd/ops/hip/` when the input lives on a ROCm device. If the compiled
extension is missing on a GPU machine the op raises instead of silently
falling back to eager (set CODA_AMD_ALLOW_EAGER=1 to override, e.g. for
kernel-vs-eager numerics comparisons on the GPU).

Cheap glue ops (argmax, masks, prior construction) run as eager
PyTorch-ROCm everywhere.
"""
from __future__ import annotations

import os

import torch

from . import reference
from ..util import DEBUG, _check
from .reference import (  # re-export cheap ops + constants
    PBEST_NUM_POINTS, GRID_LO, GRID_HI, EPS_PROB, LOG_CLAMP,
    consensus, confusion_prior, init_dirichlets, dirichlet_to_beta,
    pi_hat_partial, pi_hat_pack, pi_hat_pack_chunked,
    pi_hat_partial_packed, pi_hat_partial_streamed, pi_hat_normalize,
    pbest_from_beta_hchunked, init_model_stats,
    beta_grid_pdf_cdf, hypothetical_betas,
    mixture_entropy as _mixture_entropy_eager,
    pred_classes, disagreement_mask,
    accuracy_losses, entropy_acquisition, vma_pairwise, lure_weights,
)

_ext = None
_ext_err = None


def _load_ext():
    """Load the in-tree HIP extension (coda_amd/ops/_coda_hip*.so)."""
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from . import _coda_hip  # noqa: F401  (built by build_hip.py, in-tree)
        _ext = _coda_hip
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _ext_err = e
    return _ext


def hip_available() -> bool:
    return _load_ext() is not None


def _want_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if _load_ext() is not None:
        return True
    if os.environ.get("CODA_AMD_ALLOW_EAGER") == "1":
        return False
    raise RuntimeError(
        "coda_amd HIP extension is not built but input is on a ROCm device. "
        "Run `python build_hip.py` (or __graft_entry__.build()) to compile "
        f"the gfx950 kernels. Import error: {_ext_err}")


def pbest_from_beta(alpha: torch.Tensor, beta: torch.Tensor,
                    num_points: int = PBEST_NUM_POINTS,
                    return_unnormalized: bool = False) -> torch.Tensor:
    """P(model h is best) per row from diagonal Beta params: (R,H) -> (R,H).

    The HIP kernel is compiled for the reference's P=256 grid; other grid
    sizes run the eager formulation (on GPU too - that is a shape choice,
    not a silent fallback).
    """
    H = alpha.shape[-1]
    if H > 2048:  # beyond the fused kernel's LDS budget
        assert not return_unnormalized
        if num_points == PBEST_NUM_POINTS and _want_hip(alpha):
            out = _pbest_wide_hip(alpha, beta)
            if DEBUG:
                _check(out, "pbest(wide)")
            return out
        return reference.pbest_from_beta_hchunked(alpha, beta, num_points)
    if (num_points == PBEST_NUM_POINTS and not return_unnormalized
            and _want_hip(alpha)):
        out = _ext.pbest_from_beta(alpha.contiguous(), beta.contiguous(),
                                   int(num_points))
        if DEBUG:
            _check(out, "pbest(kernel)")
        return out
    return reference.pbest_from_beta(alpha, beta, num_points,
                                     return_unnormalized)


def _pbest_wide_hip(alpha: torch.Tensor, beta: torch.Tensor,
                    hc: int = 0) -> torch.Tensor:
    """Wide-H P(best) through the two-pass window kernels.

    The LDS-window chunks play the role of ranks in the sharded math
    (ops/sharded.py): pass-A slog2 partials sum to the global H-coupling,
    pass B integrates each window against it. Identical math to the
    fused kernel, two launches + two tiny reductions total.

    hc (window size, <= 2048 for LDS) defaults to the largest window
    that still yields >= ~512 workgroups - few rows (the per-class
    posterior refresh is R = C rows) need narrow windows to fill
    256 CUs.
    """
    R, H = alpha.shape
    if hc <= 0:
        bx = max(1, (R + 3) // 4)
        kh_needed = max(1, (512 + bx - 1) // bx)
        hc = min(512, max(64, (H + kh_needed - 1) // kh_needed))
    a, b = alpha.contiguous(), beta.contiguous()
    slog2 = _ext.pbest_phase1_wide(a, b, hc).sum(0)
    pb, tot_part = _ext.pbest_phase2_wide(a, b, slog2.contiguous(), hc)
    tot = tot_part.sum(0)
    return pb / tot.clamp_min(reference.EPS_PROB).unsqueeze(-1)


def mixture_entropy(pbest_rows: torch.Tensor, pi_hat: torch.Tensor):
    """mixture0 (H,) + its log2 entropy (reference mixture_entropy).

    GPU: torch reduces the (C, H) column sum with a 32-thread launch
    (~22 us at C=1000, H=128); the HIP twin (mix_part/mix_combine in
    pbest.hip) does slab partials + a fused entropy contraction in
    ~8 us, deterministic fixed-order sums."""
    H = pbest_rows.size(1)
    if (pbest_rows.is_cuda and H % 4 == 0 and H <= 1024
            and _want_hip(pbest_rows)):
        m0, h0 = _ext.mixture_entropy(pbest_rows.contiguous(),
                                      pi_hat.contiguous())
        return m0, h0.reshape(())
    return _mixture_entropy_eager(pbest_rows, pi_hat)


def pi_hat_delta(preds: torch.Tensor, point_classes: torch.Tensor,
                 chunk_h: int = 32,
                 preds_t: torch.Tensor = None) -> torch.Tensor:
    """sum_h preds[h, :, cls_h] -> (N,) fp32 (the rank-1 pi_hat term).

    preds_t, when given, is the static class-major (H, C, N) mirror of
    preds: the row-major gather touches one element per 64-B sector
    (at the random-sector floor, 150 us at the headline shape); the
    mirror makes the same read a coalesced stream (~15x). The kernels
    share the accumulation pattern, so the route is bitwise-neutral."""
    if (preds.is_cuda and preds.is_contiguous()
            and preds.dtype in (torch.float32, torch.bfloat16,
                                torch.float8_e4m3fn)
            and _want_hip(preds)):
        cls32 = point_classes.to(torch.int32).contiguous()
        H, N = preds.shape[0], preds.shape[1]
        # N threads alone may not fill 256 CUs (H=128 x N=50k is 196
        # blocks): chunk H across blockIdx.y until >= 512 blocks, then
        # reduce partials deterministically on-device
        bx = (N + 255) // 256
        kh = -(-512 // bx)
        if kh > 1 and H >= 2 * kh:
            hc = -(-H // kh)
            if preds_t is not None:
                out = _ext.pi_hat_delta_t_part(preds_t, cls32, hc).sum(0)
            else:
                out = _ext.pi_hat_delta_part(preds, cls32, hc).sum(0)
        else:
            # small-N shapes keep the row-major kernel (its single-chain
            # accumulation differs bitwise from the 4-acc part kernels)
            out = _ext.pi_hat_delta(preds, cls32)
        if DEBUG:
            _check(out, "pi_hat_delta(kernel)")
        return out
    return reference.pi_hat_delta(preds, point_classes, chunk_h)


def eig_chunk(alpha_cc: torch.Tensor, beta_cc: torch.Tensor,
              chunk_classes: torch.Tensor, pbest_before: torch.Tensor,
              pi_hat: torch.Tensor, pi_hat_xi_chunk: torch.Tensor,
              mixture0: torch.Tensor, H_before: torch.Tensor,
              update_weight: float = 1.0,
              num_points: int = PBEST_NUM_POINTS) -> torch.Tensor:
    """Fused per-chunk EIG: hypothetical Beta updates -> P(best) rows -> EIG.

    alpha_cc/beta_cc: (H, C) current diagonal Betas.
    chunk_classes: (B, H) argmax class of each model on each candidate.
    pbest_before: (C, H); pi_hat: (C,); pi_hat_xi_chunk: (B, C);
    mixture0: (H,); H_before: () scalar tensor.  Returns eig: (B,).

    Reference semantics: coda/coda.py:261-278 (K10 + K6-K9 + K11).
    """
    if num_points == PBEST_NUM_POINTS and _want_hip(alpha_cc):
        out = _ext.eig_chunk(
            alpha_cc.contiguous(), beta_cc.contiguous(),
            chunk_classes.to(torch.int32).contiguous(),
            pbest_before.contiguous(), pi_hat.contiguous(),
            pi_hat_xi_chunk.contiguous(), mixture0.contiguous(),
            float(H_before), float(update_weight), int(num_points))
        if DEBUG:
            _check(out, "eig(kernel)")
        return out
    a, b = reference.hypothetical_betas(alpha_cc, beta_cc, chunk_classes,
                                        update_weight)
    B, C, H = a.shape
    pbest_hyp = reference.pbest_from_beta(
        a.reshape(B * C, H), b.reshape(B * C, H), num_points).reshape(B, C, H)
    return reference.eig_assemble(pbest_hyp, pbest_before, pi_hat,
                                  pi_hat_xi_chunk, mixture0, H_before)
