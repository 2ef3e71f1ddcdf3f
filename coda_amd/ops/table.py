"""Table-factored EIG (v2): deduplicated Beta curves + MFMA GEMM pairing.

The fused kernel (v1) evaluates, per candidate row (b, c), H Beta
pdf/cdf curves - but across a chunk there are only TWO possible curves
per (class c, model h): the hypothetical update either bumps alpha
(model h predicted c) or beta (it predicted something else). That is
C*H*2 distinct curves versus B*C*H evaluated - a factor B/2 (~128x at
B=256) of redundant transcendental work.

v2 factors the whole chunk through per-step tables:

  1. table_precompute: all C*H*2 curves once ->
       EG[c,h,v,p]  = 2^(log2 pdf - log2 cdf)      (the per-model factor)
       Delta[c,h,p] = log2 cdf_v1 - log2 cdf_v0
       S_base[c,p]  = sum_h log2 cdf_v0
  2. per chunk: slog[b,c,p] = S_base[c] + sum_{h: cls(b,h)=c} Delta[c,h]
     (one scatter-add - each model contributes to exactly one class),
     ES = 2^slog * trapz_weights.
  3. the P(best) integrals become, for each class c, ONE GEMM
       M[c] = EG[c] (2H x P)  @  ES[c] (P x B)
     i.e. a (C, 2H, P) x (C, P, B) batched GEMM on the matrix cores,
     followed by a (B, C, H) gather selecting v = [cls(b,h) == c].

Numerics: identical math to v1 up to the order of the +-80-log-clamp
(which never binds on the + side because log2 cdf >= log2(1e-30) =
-99.66 > -115.4, and both formulations underflow identically on the -
side); pdf log-arguments are f64-anchored exactly like the kernel.

Everything is dense fp32 tensor work: on MI355X the GEMM runs on MFMA
via rocBLAS and the rest is bandwidth-bound elementwise - no
per-candidate transcendentals at all.
"""
from __future__ import annotations

from typing import NamedTuple

import torch

from .reference import (EPS_PROB, GRID_HI, GRID_LO, PBEST_NUM_POINTS,
                        eig_assemble)

_LOG2E = 1.4426950408889634


def _use_bf16_gemm() -> bool:
    import os
    return os.environ.get("CODA_AMD_V2_GEMM", "bf16") != "fp32"


def _class_csr(cls: torch.Tensor, C: int):
    """Sort each candidate's models by predicted class -> CSR buckets.

    cls: (B, H). Returns (hvals (B,H) int32 - model indices sorted by
    class; offsets (B, C+1) int32). int32 sort + batched searchsorted
    (an int64 sort + a (B,H) scatter_add cost 0.57 ms/step at the
    10k-model pool).
    """
    B, H = cls.shape
    sorted_cls, order = torch.sort(cls.to(torch.int32), dim=1,
                                   stable=True)
    bounds = torch.arange(C + 1, dtype=torch.int32,
                          device=cls.device).expand(B, C + 1)
    offsets = torch.searchsorted(sorted_cls, bounds).to(torch.int32)
    return order.to(torch.int32).contiguous(), offsets.contiguous()


class EigTables(NamedTuple):
    EG: torch.Tensor       # (C, H, 2, P)
    delta: torch.Tensor    # (C, H, P)
    s_base: torch.Tensor   # (C, P)
    weights: torch.Tensor  # (P,) trapz weights * dx
    # bf16 mirror of EG reshaped (C, 2H, P) for the GEMM fast path (GPU
    # only; bf16 inputs + f32 accumulate move the pairing to the bf16
    # MFMA rate - measured EIG impact ~5e-7 absolute, at the level of
    # fp32 reduction-order noise). CODA_AMD_V2_GEMM=fp32 disables.
    eg16: torch.Tensor = None
    # pair-engine (v3) operands, built by ops/pair.py
    # attach_pair_tables: egw (C, 2H, P) bf16 = the pairing-GEMM B
    # operand with baseline curve + trapz weights folded in; delta16
    # (C, H, P) fp16 = the dsum kernel's halved-traffic delta table.
    egw: torch.Tensor = None
    delta16: torch.Tensor = None
    # (C, P) fp32 per-class delta totals (complement-segment dsum)
    dall: torch.Tensor = None


def table_precompute(alpha_cc: torch.Tensor, beta_cc: torch.Tensor,
                     update_weight: float = 1.0,
                     num_points: int = PBEST_NUM_POINTS) -> EigTables:
    """Build the per-step curve tables from the (H, C) diagonal Betas.

    Chunked over class rows: the naive form holds ~6 (C*H*2, P) fp32
    transients at once (~120 GB at a 10k-model pool), which OOMs next
    to a resident prediction tensor. Output tables are allocated once
    and filled per chunk (a few GB of transients regardless of H*C).
    """
    H, C = alpha_cc.shape
    dev = alpha_cc.device
    P = num_points

    x = torch.linspace(GRID_LO, GRID_HI, P, device=dev,
                       dtype=torch.float64)
    lx = torch.log2(x)
    l1mx = torch.log2(1.0 - x)
    dx = float(x[1] - x[0])

    EG = torch.empty(C, H, 2, P, device=dev)
    delta = torch.empty(C, H, P, device=dev)
    s_base = torch.empty(C, P, device=dev)
    dall = torch.empty(C, P, device=dev) if EG.is_cuda else None
    eg16 = None
    if EG.is_cuda and _use_bf16_gemm():
        eg16 = torch.empty(C, 2 * H, P, device=dev, dtype=torch.bfloat16)

    # rows per chunk so each (rows*H*2, P) fp32 transient stays <= ~1 GB
    cchunk = max(1, min(C, int(1e9 // (H * 2 * P * 4))))
    for c0 in range(0, C, cchunk):
        c1 = min(C, c0 + cchunk)
        # variants: v0 = (a, b+w) "predicted another class", v1 = (a+w, b)
        a = torch.stack([alpha_cc[:, c0:c1],
                         alpha_cc[:, c0:c1] + update_weight], dim=-1)
        b = torch.stack([beta_cc[:, c0:c1] + update_weight,
                         beta_cc[:, c0:c1]], dim=-1)        # (H, cc, 2)
        aR = a.permute(1, 0, 2).reshape(-1)  # (cc*H*2,) in (c, h, v)
        bR = b.permute(1, 0, 2).reshape(-1)
        lnB2 = (torch.lgamma(aR.double()) + torch.lgamma(bR.double())
                - torch.lgamma((aR + bR).double())) * _LOG2E
        # f64 log-pdf (as in the kernel's anchored form), then fp32
        t2 = ((aR.double() - 1.0).unsqueeze(-1) * lx
              + (bR.double() - 1.0).unsqueeze(-1) * l1mx
              - lnB2.unsqueeze(-1)).float()                 # (R, P)
        pdf = torch.exp2(t2)
        csum = torch.cat(
            [torch.zeros_like(pdf[:, :1]),
             torch.cumsum(0.5 * (pdf[:, 1:] + pdf[:, :-1]) * dx,
                          dim=-1)], dim=-1)
        lc = torch.log2(csum.clamp_min(EPS_PROB))           # (R, P)
        cc = c1 - c0
        EG[c0:c1] = torch.exp2(t2 - lc).reshape(cc, H, 2, P)
        lc = lc.reshape(cc, H, 2, P)
        delta[c0:c1] = lc[:, :, 1] - lc[:, :, 0]
        s_base[c0:c1] = lc[:, :, 0].sum(dim=1)
        if dall is not None:
            dall[c0:c1] = delta[c0:c1].sum(dim=1)
        if eg16 is not None:
            eg16[c0:c1] = EG[c0:c1].reshape(cc, 2 * H, P) \
                .to(torch.bfloat16)

    w = torch.full((P,), dx, device=dev)
    w[0] = w[-1] = 0.5 * dx
    return EigTables(EG, delta, s_base, w, eg16, dall=dall)


def pbest_hyp_table(tables: EigTables,
                    chunk_classes: torch.Tensor) -> torch.Tensor:
    """Normalized hypothetical P(best): (B, H) classes -> (B, C, H)."""
    EG, delta, s_base, w = tables.EG, tables.delta, tables.s_base, tables.weights
    C, H, _, P = EG.shape
    B = chunk_classes.shape[0]
    cls = chunk_classes.long()

    # slog[b, c] = s_base[c] + sum_{h: cls(b,h)==c} delta[c, h]
    flat = delta.permute(1, 0, 2).reshape(H * C, P)         # (h, c) rows
    sel = flat[(torch.arange(H, device=cls.device) * C).unsqueeze(0)
               + cls]                                        # (B, H, P)
    slog = s_base.unsqueeze(0).repeat(B, 1, 1)              # (B, C, P)
    slog.scatter_add_(1, cls.unsqueeze(-1).expand(B, H, P), sel)

    ES = torch.exp2(slog) * w                               # (B, C, P)
    # stride-aware bmm: (P, B) columns are contiguous in p, so rocBLAS
    # consumes the permuted view directly (no 262 MB transposed copy)
    M = torch.bmm(EG.reshape(C, H * 2, P),
                  ES.permute(1, 2, 0))                      # (C, 2H, B)
    Mv = M.view(C, H, 2, B)
    # v-selection: v=1 only where cls(b,h) == c - exactly H of the C*H
    # positions per candidate. Start from the v=0 plane and scatter the
    # H hit values per candidate (tiny) instead of a dense (B,C,H,2)
    # gather.
    pb = Mv[:, :, 0, :].permute(2, 0, 1).contiguous()       # (B, C, H)
    ar_h = torch.arange(H, device=cls.device)
    ar_b = torch.arange(B, device=cls.device)
    hit_vals = Mv[cls, ar_h.view(1, H), 1, ar_b.view(B, 1)]  # (B, H)
    pb[ar_b.view(B, 1), cls, ar_h.view(1, H)] = hit_vals
    return pb / pb.sum(-1, keepdim=True).clamp_min(EPS_PROB)


def eig_chunk_table(tables: EigTables, chunk_classes: torch.Tensor,
                    pbest_before: torch.Tensor, pi_hat: torch.Tensor,
                    pi_hat_xi_chunk: torch.Tensor, mixture0: torch.Tensor,
                    H_before: torch.Tensor) -> torch.Tensor:
    """(B,) EIG for a candidate chunk through the v2 tables.

    GPU path: the es_build / eig_assemble_k fusion kernels around the
    batched GEMM (see pbest.hip); CPU path: the torch composition above.
    """
    EG, delta, s_base, w = tables.EG, tables.delta, tables.s_base, tables.weights
    C, H, _, P = EG.shape
    import coda_amd.ops as O
    if EG.is_cuda and P == PBEST_NUM_POINTS and O._want_hip(EG):
        cls32 = chunk_classes.to(torch.int32).contiguous()
        hvals, offsets = _class_csr(chunk_classes.long(), C)
        bf16 = tables.eg16 is not None
        ES = O._ext.es_build(s_base, delta, tables.dall, hvals,
                             offsets, w, bf16)
        if bf16:
            M = torch.bmm(ES, tables.eg16.transpose(1, 2))
        else:
            M = torch.bmm(ES, EG.reshape(C, 2 * H, P).transpose(1, 2))
        h_after = O._ext.eig_assemble_k(M, cls32, pi_hat.contiguous(),
                                        pbest_before.contiguous(),
                                        mixture0.contiguous())  # (B, C)
        return H_before - (pi_hat_xi_chunk * h_after).sum(-1)
    pb = pbest_hyp_table(tables, chunk_classes)
    return eig_assemble(pb, pbest_before, pi_hat, pi_hat_xi_chunk,
                        mixture0, H_before)


def table_update_rows(tables: EigTables, alpha_cc: torch.Tensor,
                      beta_cc: torch.Tensor, rows,
                      update_weight: float = 1.0) -> EigTables:
    """Recompute the table slices for the given class rows in place.

    add_label moves only Dirichlet row `true_class` of each model
    (coda/coda.py:316-317), so between steps only that class's H*2
    curves change - an O(1/C) refresh instead of a full rebuild.
    """
    rows = torch.as_tensor(rows, device=alpha_cc.device, dtype=torch.long)
    if rows.numel() == 0:
        return tables
    import coda_amd.ops as O
    if (alpha_cc.is_cuda and tables.EG.shape[-1] == PBEST_NUM_POINTS
            and rows.numel() <= 4 and O._want_hip(alpha_cc)):
        if (tables.eg16 is not None and tables.egw is not None
                and tables.delta16 is not None
                and tables.dall is not None):
            # fully-fused per-class commit (curves + sums + all table
            # writes in 3 kernels - the torch chain below spends
            # ~50 us/class in 32-thread strided reductions and small
            # conversion launches on the eager/distributed path)
            for c in rows.tolist():
                O._ext.table_commit_cols(
                    alpha_cc[:, c].contiguous(),
                    beta_cc[:, c].contiguous(), int(c),
                    tables.EG, tables.delta, tables.s_base,
                    tables.weights, tables.eg16, tables.egw,
                    tables.delta16, tables.dall, float(update_weight))
            return tables
        # per-class refresh kernel: one wave per (model, variant) curve
        H = alpha_cc.shape[0]
        for c in rows.tolist():
            eg, lc = O._ext.beta_row_tables(
                alpha_cc[:, c].contiguous(), beta_cc[:, c].contiguous(),
                float(update_weight))
            tables.EG[c] = eg
            if tables.eg16 is not None:
                tables.eg16[c] = eg.reshape(2 * H, -1).to(torch.bfloat16)
            tables.delta[c] = lc[:, 1] - lc[:, 0]
            tables.s_base[c] = lc[:, 0].sum(0)
            if tables.dall is not None:
                tables.dall[c] = tables.delta[c].sum(0)
        if tables.egw is not None:
            from .pair import update_egw_rows
            update_egw_rows(tables, rows.tolist())
        return tables
    sub = table_precompute(alpha_cc[:, rows], beta_cc[:, rows],
                           update_weight, tables.EG.shape[-1])
    tables.EG[rows] = sub.EG
    if tables.eg16 is not None:
        H = alpha_cc.shape[0]
        tables.eg16[rows] = sub.EG.reshape(
            rows.numel(), 2 * H, -1).to(torch.bfloat16)
    tables.delta[rows] = sub.delta
    tables.s_base[rows] = sub.s_base
    if tables.dall is not None:
        tables.dall[rows] = tables.delta[rows].sum(1)
    if tables.egw is not None:
        from .pair import update_egw_rows
        update_egw_rows(tables, rows.tolist())
    return tables


# ---------------------------------------------------------------------------
# Sharded v2: the model axis is split across ranks; each rank holds table
# slices for its local models. The cross-rank coupling is carried by an
# all-gather of the SELECTED delta curves - (B, H, P) floats per chunk,
# world-times smaller than all-reducing the dense (B, C, P) slog - plus
# a once-per-step (C, P) all-reduce for the base term and two tiny (B, C)
# all-reduces (normalizer, entropy partials).
# ---------------------------------------------------------------------------

def s_base_global(tables: EigTables, comm) -> torch.Tensor:
    """sum over ALL models of log2 cdf_v0: all-reduced (C, P)."""
    out = tables.s_base.clone()
    comm.all_reduce_(out)
    return out


def eig_chunk_table_sharded(tables: EigTables, s_base_all: torch.Tensor,
                            cls_local: torch.Tensor,
                            pbest_before_local: torch.Tensor,
                            pi_hat: torch.Tensor,
                            pi_hat_xi_chunk: torch.Tensor,
                            mixture0_local: torch.Tensor,
                            H_before: torch.Tensor, comm,
                            hsizes=None) -> torch.Tensor:
    """(B,) EIG with local tables; identical on every rank.

    hsizes: per-rank model counts (comm.shard_sizes(H_global)) - lets the
    per-chunk gathers skip their size-exchange collectives."""
    EG, delta, w = tables.EG, tables.delta, tables.weights
    C, Hl, _, P = EG.shape
    B = cls_local.shape[0]
    cls_l = cls_local.long()

    flat = delta.permute(1, 0, 2).reshape(Hl * C, P)
    sel = flat[(torch.arange(Hl, device=cls_l.device) * C).unsqueeze(0)
               + cls_l].contiguous()                         # (B, Hl, P)
    sel_all = comm.all_gather_cat(sel, dim=1,
                                  sizes=hsizes).contiguous()  # (B, H, P)
    cls_all = comm.all_gather_cat(cls_l, dim=1, sizes=hsizes)  # (B, H)
    Hg = cls_all.shape[1]

    import coda_amd.ops as O
    if EG.is_cuda and P == PBEST_NUM_POINTS and O._want_hip(EG):
        cls_l32 = cls_l.to(torch.int32).contiguous()
        hvals, offsets = _class_csr(cls_all, C)
        bf16 = tables.eg16 is not None
        ES = O._ext.es_build_gathered(s_base_all.contiguous(), sel_all,
                                      hvals, offsets, w, bf16)
        if bf16:
            M = torch.bmm(ES, tables.eg16.transpose(1, 2))
        else:
            M = torch.bmm(ES, EG.reshape(C, 2 * Hl, P).transpose(1, 2))
        tot = O._ext.eig_totals(M, cls_l32)                  # (B, C) partial
        comm.all_reduce_(tot)
        h_after = O._ext.eig_entropy(M, cls_l32, tot.contiguous(),
                                     pi_hat.contiguous(),
                                     pbest_before_local.contiguous(),
                                     mixture0_local.contiguous())
        comm.all_reduce_(h_after)
        return H_before - (pi_hat_xi_chunk * h_after).sum(-1)

    slog = s_base_all.unsqueeze(0).repeat(B, 1, 1)           # (B, C, P)
    slog.scatter_add_(1, cls_all.unsqueeze(-1).expand(B, Hg, P), sel_all)
    ES = torch.exp2(slog) * w

    M = torch.bmm(EG.reshape(C, Hl * 2, P),
                  ES.permute(1, 2, 0))                       # (C, 2Hl, B)
    Mv = M.view(C, Hl, 2, B)
    pb = Mv[:, :, 0, :].permute(2, 0, 1).contiguous()        # (B, C, Hl)
    ar_h = torch.arange(Hl, device=cls_l.device)
    ar_b = torch.arange(B, device=cls_l.device)
    hit_vals = Mv[cls_l, ar_h.view(1, Hl), 1, ar_b.view(B, 1)]  # (B, Hl)
    pb[ar_b.view(B, 1), cls_l, ar_h.view(1, Hl)] = hit_vals
    tot = pb.sum(-1)                                         # (B, C)
    comm.all_reduce_(tot)
    pb = pb / tot.clamp_min(EPS_PROB).unsqueeze(-1)

    d = pi_hat.view(1, C, 1) * (pb - pbest_before_local.unsqueeze(0))
    m = (mixture0_local.view(1, 1, Hl) + d).clamp_min(1e-12)
    H_after = -(m * m.log2()).sum(-1)                        # (B, C) partial
    comm.all_reduce_(H_after)
    return H_before - (pi_hat_xi_chunk * H_after).sum(-1)
