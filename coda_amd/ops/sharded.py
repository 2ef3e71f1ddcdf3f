"""Model-axis-sharded variants of the hot ops.

Each rank holds Beta params for its local slice of models; the coupling
across models (the exclusive-product term of the P(best) integral and the
entropy sums of EIG) crosses shards through three small all-reduces per
chunk:
  1. sum_h log cdf_h on the grid        -> (R, P)
  2. the P(best) normalizer             -> (R,)
  3. partial mixture entropies          -> (B, C)

With world == 1 these degrade to the single-device reference math (the
all-reduces are no-ops) and are used by the CPU tests to check
shard-vs-single equivalence.
"""
from __future__ import annotations

import torch

from . import reference
from .reference import EPS_PROB, LOG_CLAMP, PBEST_NUM_POINTS


def _hip(t: torch.Tensor, num_points: int):
    """HIP two-phase path availability (loud failure on GPU handled by
    ops._want_hip)."""
    import coda_amd.ops as O
    return O._want_hip(t) and num_points == PBEST_NUM_POINTS


def pbest_from_beta_sharded(alpha_local: torch.Tensor,
                            beta_local: torch.Tensor, comm,
                            num_points: int = PBEST_NUM_POINTS):
    """P(best) for the LOCAL models, with the H-coupling all-reduced.

    alpha_local, beta_local: (R, H_local).
    Returns prob_local: (R, H_local), normalized over the GLOBAL model axis.

    GPU path: the two-phase gfx950 kernels with the RCCL all-reduce of the
    (R, P) log-cdf partials between pass A and pass B, plus the tiny (R,)
    normalizer all-reduce. All ranks must use the same path (the eager
    partial is in natural log, the HIP partial in log2).
    """
    if _hip(alpha_local, num_points):
        import coda_amd.ops as O
        a = alpha_local.contiguous()
        b = beta_local.contiguous()
        slog2 = O._ext.pbest_phase1(a, b)
        comm.all_reduce_(slog2)
        pb, tot = O._ext.pbest_phase2(a, b, slog2)
        comm.all_reduce_(tot)
        return pb / tot.clamp_min(EPS_PROB).unsqueeze(-1)
    pdf, cdf, x = reference.beta_grid_pdf_cdf(alpha_local, beta_local,
                                              num_points)
    log_cdf = torch.log(cdf.clamp_min(EPS_PROB))          # (R, Hl, P)
    slog = log_cdf.sum(dim=1)                             # (R, P)
    comm.all_reduce_(slog)                                # global sum over H
    prod_excl = torch.exp(
        (slog.unsqueeze(1) - log_cdf).clamp(-LOG_CLAMP, LOG_CLAMP))
    prob = torch.trapz(pdf * prod_excl, x, dim=-1)        # (R, Hl)
    total = prob.sum(-1)                                  # (R,)
    comm.all_reduce_(total)
    return prob / total.clamp_min(EPS_PROB).unsqueeze(-1)


def eig_chunk_sharded(alpha_cc_local: torch.Tensor, beta_cc_local: torch.Tensor,
                      chunk_classes_local: torch.Tensor,
                      pbest_before_local: torch.Tensor, pi_hat: torch.Tensor,
                      pi_hat_xi_chunk: torch.Tensor, mixture0_local: torch.Tensor,
                      H_before: torch.Tensor, comm,
                      update_weight: float = 1.0,
                      num_points: int = PBEST_NUM_POINTS) -> torch.Tensor:
    """Sharded fused EIG chunk; every rank returns the identical (B,) EIG.

    alpha_cc_local/beta_cc_local: (H_local, C); chunk_classes_local:
    (B, H_local); pbest_before_local: (C, H_local); mixture0_local:
    (H_local,) - this rank's slice of the global mixture.
    """
    if _hip(alpha_cc_local, num_points):
        import coda_amd.ops as O
        Hl, C = alpha_cc_local.shape
        B = chunk_classes_local.shape[0]
        cls = chunk_classes_local.to(torch.int32).contiguous()
        slog2 = O._ext.eig_phase1(alpha_cc_local.contiguous(),
                                  beta_cc_local.contiguous(), cls,
                                  float(update_weight))
        comm.all_reduce_(slog2)
        pb, tot = O._ext.eig_phase2(alpha_cc_local.contiguous(),
                                    beta_cc_local.contiguous(), cls, slog2,
                                    float(update_weight))
        comm.all_reduce_(tot)
        pbest_hyp = (pb / tot.clamp_min(EPS_PROB).unsqueeze(-1)) \
            .view(B, C, Hl)
    else:
        a, b = reference.hypothetical_betas(alpha_cc_local, beta_cc_local,
                                            chunk_classes_local,
                                            update_weight)
        B, C, Hl = a.shape
        pbest_hyp = pbest_from_beta_sharded(
            a.reshape(B * C, Hl), b.reshape(B * C, Hl), comm,
            num_points).reshape(B, C, Hl)
    deltas = pi_hat.view(1, C, 1) * (pbest_hyp - pbest_before_local.unsqueeze(0))
    mix_new = mixture0_local.view(1, 1, Hl) + deltas
    m = mix_new.clamp_min(1e-12)
    H_after = -(m * m.log2()).sum(-1)                     # (B, C) partial
    comm.all_reduce_(H_after)                             # sum over global H
    return H_before - (pi_hat_xi_chunk * H_after).sum(-1)


def mixture_entropy_sharded(pbest_rows_local: torch.Tensor,
                            pi_hat: torch.Tensor, comm):
    """mixture0 over local models + globally-summed log2 entropy.

    pbest_rows_local: (C, H_local) -> (mixture0_local (H_local,), H0 scalar).
    """
    mixture0 = (pi_hat.unsqueeze(-1) * pbest_rows_local).sum(0)
    m = mixture0.clamp_min(1e-12)
    H0 = -(m * m.log2()).sum().reshape(1)
    comm.all_reduce_(H0)
    return mixture0, H0[0]
