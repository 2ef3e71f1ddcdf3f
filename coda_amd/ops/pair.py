"""Pair-factored EIG (v3): hit-sparse acquisition over the full pool.

The canonical acquisition (reference main.py:49 `--prefilter-n 0`,
hot loop coda/coda.py:235-281) scores EVERY disagreeing unlabeled point
per step. The v2 table engine (ops/table.py) still does dense work per
(candidate, class) cell: a (B, C, P) ES tensor + a (C, 2H, P) x (C, P, B)
GEMM - ~51 GB of ES traffic per step at the headline config (B=50k,
C=1000, P=256), 85.9 ms/step measured in round 1.

v3 exploits the hit sparsity: for candidate b and hypothesized class c,
the hypothetical update only deviates from the label-free baseline when
at least one model PREDICTS class c on point b (coda/coda.py:150-168:
alpha bumps only for argmax==c models). A candidate's H models hit at
most min(H, C) distinct classes - typically ~5-40 - so of the B*C cells
only K = sum_b |distinct classes(b)| ~ B*30 are non-baseline:

    EIG[b] = H_before - sum_c pi_xi[b,c] * h_base[c]
                      - sum_{c in hit(b)} pi_xi[b,c] * (h_after[b,c] - h_base[c])

with h_base[c] the entropy of the baseline (no-hit) hypothetical - one
value per class per step. The per-pair work is one P-point curve
(exp2 of the summed log-cdf deltas) and one (P,) x (P, 2H) pairing GEMV
against the class's curve table, grouped by class for MFMA tiling
(ops/hip/pair.hip). ~200x less arithmetic than the dense form at the
headline shape, and the per-step launch count drops from ~6 per
256-candidate chunk (~1200) to ~10 total.

The hit structure depends only on the cached argmax classes, so it is
STATIC across steps: built once over all disagreeing points, reused
every acquisition (labeled points are skipped at gather time).

Sharding (model axis on disk, candidate axis in compute): every rank
keeps a replicated (H, C) diagonal-Beta view (updated locally from the
init-time gathered global classes), computes v3 EIG for a strided slice
of the candidates, and all-gathers the (B,) EIG values - per-step wire
is O(B) bytes instead of the 268 MB/chunk curve gather of the v2
sharded path (ops/table.py:256 in round 1).
"""
from __future__ import annotations

from typing import NamedTuple, Optional

import torch

from .reference import EPS_PROB, PBEST_NUM_POINTS

PAIR_TILE = 16  # minimum tile; build_pairs may pad to 64 (see below)


def tile_for(H: int, B: int) -> int:
    """MFMA tile height (pairs per class-uniform tile).

    128-pair tiles divide the dominant GEMM B traffic by 8 vs 16-pair
    tiles (H <= 144: B resident in LDS, fused epilogue; larger H: the
    split wide pipeline with a 128x128 M-writing GEMM). They only pay
    off when the padding (<= tile-1 pairs per class) is small against
    the real pair count - small candidate sets use 16-pair tiles.
    """
    return 128 if B >= 4096 else PAIR_TILE


class PairStructure(NamedTuple):
    """Static hit structure for a fixed candidate set.

    Pairs are grouped by class; each class run is [base pair] + real
    pairs + padding to a `tile` multiple, so one kernel tile never
    straddles classes. Base/pad pairs carry pair_b = -1 (empty model
    segment -> they evaluate the class baseline h_base[c]).
    """
    cand_ids: torch.Tensor   # (B,) long — point ids covered
    pair_b: torch.Tensor     # (K,) int32 — row into cand_ids, -1 = base/pad
    pair_c: torch.Tensor     # (K,) int32 — hypothesized class
    seg_off: torch.Tensor    # (K+1,) int32 — segment offsets into seg_h
    seg_h: torch.Tensor      # (S,) int32 — models with argmax c on point b
    base_pos: torch.Tensor   # (C,) long — position of class c's base pair
    n_real: int              # UNIQUE evaluated pair count (post-dedupe)
    tile: int                # pairs per kernel tile (16 or 128)
    cand_off: torch.Tensor = None    # (B+1,) int32 — per-candidate CSR
    # (total hits,) int32 — representative pair position for every
    # original (candidate, class) hit, grouped by candidate
    cand_pairs: torch.Tensor = None
    # (K, ceil(H/32)) int32 — bit h set iff model h hits pair k (the
    # static v-select; reading cls rows per pair costs 1.6 ms/step at
    # the headline shape, the bitmask 0.05 ms)
    vmask: torch.Tensor = None
    # (K,) int32 — 1 where the stored segment is the COMPLEMENT of the
    # hit set (majority pairs: most models predict the true-ish label,
    # so summing the few non-hitting models' deltas and subtracting
    # from the per-class total nearly halves the dsum read traffic)
    pair_neg: torch.Tensor = None
    # (total hits,) int64 — cand_pairs with the pair's class packed in
    # the low 32 bits ((k << 32) | c): the finalize kernel's gather
    # chase drops from 3 dependent loads to 2 (PMC: wait/busy 41.6)
    cand_ck: torch.Tensor = None
    # (G+1,) int32 — 128-pair TILE groups: runs of same-class tiles
    # capped at GRP_MAX, so the GEMM's 131 KB B stage amortizes over a
    # class's tiles instead of re-staging per tile (classes average
    # ~2.4 tiles at the headline distribution)
    grp_off: torch.Tensor = None

    @property
    def K(self) -> int:
        return self.pair_b.shape[0]


def build_pairs(cls_rows: torch.Tensor, cand_ids: torch.Tensor,
                C: int, tile: int = 0,
                with_vmask: bool = True,
                dedupe: bool = True) -> PairStructure:
    """Build the hit structure from candidate argmax classes.

    cls_rows: (B, H) — argmax class of every (global) model on each
    candidate; cand_ids: (B,) point ids. All work is batched torch ops
    (one sort over B*H keys), so the one-off cost at N=50k, H=128 is a
    few ms on device. tile overrides the MFMA tile height (benchmarks).

    dedupe: h_after depends only on (class, hit SET), never on the
    candidate - and hit sets repeat massively (singletons are bounded
    by C*H across the whole pool). Duplicate sets collapse to one
    evaluated pair; the per-candidate CSR maps every original (b, c)
    hit to its representative. ~5-10x fewer evaluated pairs at the
    headline shape. Set identity via two independent 64-bit random
    per-model sums (collision odds ~K^2/2^128) plus (class, set size).
    """
    device = cls_rows.device
    B, H = cls_rows.shape
    tile = tile or tile_for(H, B)
    cls_l = cls_rows.long()
    # sort all (b, h) entries by (class, candidate) so pairs come out
    # grouped by class, and each pair's model segment is contiguous
    keys = (cls_l * B + torch.arange(B, device=device).unsqueeze(1))
    flat = keys.reshape(-1)
    order = torch.argsort(flat, stable=True)
    sk = flat[order]
    is_new = torch.ones_like(sk, dtype=torch.bool)
    is_new[1:] = sk[1:] != sk[:-1]
    pair_key = sk[is_new]                          # (K_real,)
    pr_c = pair_key // B
    pr_b = pair_key - pr_c * B
    K_real = int(pair_key.shape[0])
    pid = is_new.long().cumsum(0) - 1              # entry -> real pair
    seg_len_real = torch.bincount(pid, minlength=K_real)   # (K_real,)
    entry_h = order % H                            # (B*H,) grouped by pair

    # ---- dedupe identical (class, hit-set) pairs -------------------
    if dedupe:
        g = torch.Generator(device="cpu").manual_seed(0x5eed)
        r1 = torch.randint(-2**63, 2**63 - 1, (H,), generator=g,
                           dtype=torch.int64).to(device)
        r2 = torch.randint(-2**63, 2**63 - 1, (H,), generator=g,
                           dtype=torch.int64).to(device)
        hash1 = torch.zeros(K_real, dtype=torch.int64, device=device)
        hash2 = torch.zeros(K_real, dtype=torch.int64, device=device)
        hash1.index_add_(0, pid, r1[entry_h])      # wrapping set-sum
        hash2.index_add_(0, pid, r2[entry_h])
        ukey = torch.stack([pr_c, seg_len_real, hash1, hash2], dim=1)
        _, uid_of_real = torch.unique(ukey, dim=0, return_inverse=True)
        U = int(uid_of_real.max()) + 1 if K_real else 0
        rep_of_u = torch.full((U,), K_real, dtype=torch.int64,
                              device=device)
        rep_of_u.scatter_reduce_(0, uid_of_real,
                                 torch.arange(K_real, device=device),
                                 reduce="amin", include_self=True)
        reps_idx = torch.sort(rep_of_u).values     # ascending = (c,b) order
        new_of_u = torch.searchsorted(reps_idx, rep_of_u)  # uid -> new id
        rep_mask = torch.zeros(K_real, dtype=torch.bool, device=device)
        rep_mask[reps_idx] = True
    else:
        reps_idx = torch.arange(K_real, device=device)
        uid_of_real = reps_idx
        new_of_u = reps_idx
        rep_mask = torch.ones(K_real, dtype=torch.bool, device=device)

    pr_b2 = pr_b[reps_idx]
    pr_c2 = pr_c[reps_idx]
    seg_len2 = seg_len_real[reps_idx]
    K2 = int(reps_idx.shape[0])

    # class runs over the reduced set: base pair + reps, tile-padded
    class_counts = torch.bincount(pr_c2, minlength=C)       # (C,)
    run_len = ((class_counts + 1 + tile - 1) // tile) * tile
    run_off = torch.zeros(C + 1, dtype=torch.long, device=device)
    run_off[1:] = run_len.cumsum(0)
    K = int(run_off[-1])

    class_off = torch.zeros(C, dtype=torch.long, device=device)
    class_off[1:] = class_counts.cumsum(0)[:-1]
    rank = torch.arange(K2, device=device) - class_off[pr_c2]
    pos = run_off[:-1][pr_c2] + 1 + rank                    # (K2,)

    pair_b = torch.full((K,), -1, dtype=torch.int32, device=device)
    pair_b[pos] = pr_b2.to(torch.int32)
    pair_c = torch.repeat_interleave(
        torch.arange(C, device=device), run_len).to(torch.int32)

    # per-candidate CSR over ALL original (b, c) hits, mapped to their
    # representative's padded position (the finalize kernel's view)
    pos_of_real = pos[new_of_u[uid_of_real]]                # (K_real,)
    order_b = torch.argsort(pr_b, stable=True)
    cand_pairs = pos_of_real[order_b].to(torch.int32)
    b_counts = torch.bincount(pr_b, minlength=B)
    cand_off = torch.zeros(B + 1, dtype=torch.int32, device=device)
    cand_off[1:] = b_counts.cumsum(0).to(torch.int32)

    # entry-level views (original, pre-dedupe)
    entry_pid = pid
    flip_real = seg_len_real > (H // 2)

    # static v-select bitmask over REP pairs: bit h iff model h in the
    # TRUE hit set (before any complement rewrite). Wide pools skip it.
    vmask = None
    if with_vmask:
        W = (H + 31) // 32
        keep_v = rep_mask[entry_pid]
        tpos = pos[new_of_u[uid_of_real[entry_pid[keep_v]]]]
        hh_v = entry_h[keep_v]
        word = tpos * W + (hh_v >> 5)
        bit = torch.bitwise_left_shift(
            torch.ones_like(hh_v, dtype=torch.int32),
            (hh_v & 31).to(torch.int32))
        vmask = torch.zeros(K * W, dtype=torch.int32, device=device)
        vmask.index_put_((word,), bit, accumulate=True)
        vmask = vmask.view(K, W)

    # kernel segments: non-flipped reps store their hit models;
    # flipped (majority) reps store the COMPLEMENT - the rep
    # candidate's entries in OTHER classes - and dsum subtracts from
    # the per-class total dall[c]
    pair_neg = torch.zeros(K, dtype=torch.int32, device=device)
    mask1 = rep_mask[entry_pid] & ~flip_real[entry_pid]
    t1 = pos[new_of_u[uid_of_real[entry_pid[mask1]]]]
    h1 = entry_h[mask1]
    if bool(flip_real[reps_idx].any()):
        flipped_reps = reps_idx[flip_real[reps_idx]]        # orig pids
        pair_neg[pos[new_of_u[uid_of_real[flipped_reps]]]] = 1
        q_of_b = torch.full((B,), -1, dtype=torch.long, device=device)
        q_of_b[pr_b[flipped_reps]] = flipped_reps
        entry_b = pr_b[entry_pid]
        qb = q_of_b[entry_b]
        comp = (qb >= 0) & (entry_pid != qb)
        t2 = pos[new_of_u[uid_of_real[qb[comp]]]]
        h2 = entry_h[comp]
        tgt = torch.cat([t1, t2])
        hh = torch.cat([h1, h2])
    else:
        tgt, hh = t1, h1
    srt = torch.argsort(tgt * H + hh)                       # (tgt, h)
    seg_h = hh[srt].to(torch.int32)
    seg_len_k = torch.bincount(tgt, minlength=K)
    seg_off = torch.zeros(K + 1, dtype=torch.int32, device=device)
    seg_off[1:] = seg_len_k.cumsum(0).to(torch.int32)

    # contiguous() is load-bearing: sharded callers pass strided
    # candidate slices (ids[rank::world]), and the finalize kernel
    # walks the raw buffer - a strided view would silently read the
    # WRONG candidates' rows (caught by the GPU loopback world-4 test)
    grp_off = None
    if tile == 128 and K % 128 == 0 and K > 0:
        GRP_MAX = 4
        tc = pair_c[::128]                         # (T,) tile classes
        T = tc.numel()
        newc = torch.ones(T, dtype=torch.bool, device=device)
        newc[1:] = tc[1:] != tc[:-1]
        idx = torch.arange(T, device=device)
        run_start = idx[newc]
        run_id = newc.long().cumsum(0) - 1
        off_in_run = idx - run_start[run_id]
        gstart = newc | (off_in_run % GRP_MAX == 0)
        grp_off = torch.cat([
            idx[gstart].to(torch.int32),
            torch.tensor([T], dtype=torch.int32, device=device),
        ]).contiguous()

    return PairStructure(cand_ids=cand_ids.long().contiguous(),
                         pair_b=pair_b,
                         pair_c=pair_c, seg_off=seg_off, seg_h=seg_h,
                         base_pos=run_off[:-1].clone(), n_real=K2,
                         tile=tile, cand_off=cand_off,
                         cand_pairs=cand_pairs, vmask=vmask,
                         pair_neg=pair_neg, grp_off=grp_off,
                         cand_ck=((cand_pairs.long() << 32)
                                  | pair_c[cand_pairs.long()].long())
                         .contiguous() if cand_pairs is not None
                         else None)


def pair_h_after(tables, ps: PairStructure, cls_rows: torch.Tensor,
                 pbest_before: torch.Tensor, pi_hat: torch.Tensor,
                 mixture0: torch.Tensor) -> torch.Tensor:
    """(K,) hypothetical-update entropies, eager formulation.

    Math identical to ops/table.py's dense form restricted to hit cells
    (the kernel in ops/hip/pair.hip reproduces it in bf16-GEMM form).
    Used on CPU and as the numerics reference for the kernel; the (K,
    2H, P) gather makes it unusable at headline scale on purpose.
    """
    EG, delta, s_base, w = (tables.EG, tables.delta, tables.s_base,
                            tables.weights)
    C, H, _, P = EG.shape
    K = ps.K
    device = EG.device

    seg_len = (ps.seg_off[1:] - ps.seg_off[:-1]).long()
    seg_pair = torch.repeat_interleave(
        torch.arange(K, device=device), seg_len)
    pc_long = ps.pair_c.long()
    dsum = torch.zeros(K, P, device=device)
    if ps.seg_h.numel():
        dsum.index_add_(0, seg_pair,
                        delta[pc_long[seg_pair], ps.seg_h.long()])
    if ps.pair_neg is not None and bool(ps.pair_neg.any()):
        dall = tables.dall if getattr(tables, "dall", None) is not None \
            else delta.sum(1)
        neg = ps.pair_neg.bool()
        dsum[neg] = dall[pc_long[neg]] - dsum[neg]
    A = torch.exp2(dsum + s_base[pc_long]) * w              # (K, P)
    M = torch.einsum('kp,kjp->kj', A,
                     EG.reshape(C, 2 * H, P)[pc_long])      # (K, 2H)

    b_safe = ps.pair_b.long().clamp_min(0)
    v = (cls_rows[b_safe] == pc_long.unsqueeze(1)) \
        & (ps.pair_b >= 0).unsqueeze(1)                     # (K, H)
    j = 2 * torch.arange(H, device=device).unsqueeze(0) + v.long()
    pb = M.gather(1, j)                                     # (K, H)
    tot = pb.sum(-1, keepdim=True).clamp_min(EPS_PROB)
    pbn = pb / tot
    mm = (mixture0.unsqueeze(0)
          + pi_hat[pc_long].unsqueeze(1)
          * (pbn - pbest_before[pc_long])).clamp_min(1e-12)
    return -(mm * torch.log2(mm)).sum(-1)                   # (K,)


def eig_from_pairs(h_after: torch.Tensor, ps: PairStructure,
                   adjusted: torch.Tensor, row_sums: torch.Tensor,
                   H_before) -> torch.Tensor:
    """(N,) EIG over all points from per-(unique-)pair entropies, via
    the per-candidate CSR (each original hit maps to its
    representative pair). Points outside the candidate set get the
    baseline-only value (never read - the caller gathers cand_ids)."""
    N = adjusted.shape[0]
    h_base = h_after[ps.base_pos]                           # (C,)
    inv_rs = 1.0 / row_sums.clamp_min(1e-12)
    base_n = (adjusted @ h_base) * inv_rs                   # (N,)
    counts = (ps.cand_off[1:] - ps.cand_off[:-1]).long()
    ids = torch.repeat_interleave(ps.cand_ids, counts)      # (K_real,)
    k = ps.cand_pairs.long()
    c_v = ps.pair_c[k].long()
    pix = adjusted[ids, c_v] * inv_rs[ids]
    corr = torch.zeros(N, device=adjusted.device)
    corr.index_add_(0, ids, pix * (h_after[k] - h_base[c_v]))
    if torch.is_tensor(H_before):
        H_before = H_before.to(base_n.dtype)
    return H_before - base_n - corr


def eig_pairs(tables, ps: PairStructure, cls_rows: torch.Tensor,
              pbest_before: torch.Tensor, pi_hat: torch.Tensor,
              mixture0: torch.Tensor, H_before,
              adjusted: torch.Tensor,
              row_sums: torch.Tensor) -> torch.Tensor:
    """Full v3 EIG: (B,) values, one per structure candidate row.

    Dispatches the per-pair work to the HIP kernels when available
    (GPU), else the eager formulation.
    """
    import coda_amd.ops as O
    EG = tables.EG
    C, H, _, P = EG.shape
    if (EG.is_cuda and P == PBEST_NUM_POINTS
            and getattr(tables, "egw", None) is not None
            and O._want_hip(EG)):
        A16 = O._ext.pair_dsum_es(tables.delta16, tables.dall,
                                  ps.pair_c, ps.pair_neg, ps.seg_off,
                                  ps.seg_h)                 # (K, P) bf16
        if ps.vmask is None:
            h_after = O._ext.pair_gemm_entropy_cls(
                A16, tables.egw, ps.pair_b, ps.pair_c,
                cls_rows.to(torch.int32).contiguous(),
                pi_hat.contiguous(), pbest_before.contiguous(),
                mixture0.contiguous())                      # (K,)
        else:
            grp = (ps.grp_off if ps.grp_off is not None
                   else torch.empty(0, dtype=torch.int32,
                                    device=A16.device))
            h_after = O._ext.pair_gemm_entropy(
                A16, tables.egw, ps.vmask, ps.pair_c,
                pi_hat.contiguous(), pbest_before.contiguous(),
                mixture0.contiguous(), ps.tile, 0, grp)     # (K,)
        h_base = h_after.index_select(0, ps.base_pos).contiguous()
        q = O._ext.pair_eig_finalize(
            h_after, h_base, ps.pair_c,
            ps.cand_off, ps.cand_ck, ps.cand_ids,
            adjusted.contiguous(), row_sums.contiguous(),
            float(H_before))                                # (B,)
        return q
    h_after = pair_h_after(tables, ps, cls_rows, pbest_before,
                           pi_hat, mixture0)
    eig_n = eig_from_pairs(h_after, ps, adjusted, row_sums, H_before)
    return eig_n[ps.cand_ids]


def attach_pair_tables(tables):
    """Derive the pair-engine operands from the v2 tables:

    egw (C, 2H, P) bf16 — MFMA B operand with the baseline curve and
    trapezoid weights folded in: egw[c,j,p] = EG[c,j,p] * 2^s_base[c,p]
    * w[p] (the pair GEMM's A operand is then just 2^dsum);
    delta16 (C, H, P) fp16 — the dsum kernel's table (halves its
    traffic; |delta| < 116 fits fp16 range, and the 5e-4 relative
    rounding is below the bf16 rounding of the A operand it feeds).
    """
    EG, s_base, w = tables.EG, tables.s_base, tables.weights
    C, H, _, P = EG.shape
    esb = torch.exp2(s_base) * w                            # (C, P)
    egw = (EG.reshape(C, 2 * H, P)
           * esb.unsqueeze(1)).to(torch.bfloat16).contiguous()
    dall = tables.dall if tables.dall is not None \
        else tables.delta.sum(1).contiguous()
    return tables._replace(egw=egw,
                           delta16=tables.delta.to(torch.float16),
                           dall=dall)


def update_egw_rows(tables, rows) -> None:
    """Refresh egw/delta16 class rows (after table_update_rows)."""
    EG, s_base, w = tables.EG, tables.s_base, tables.weights
    C, H, _, P = EG.shape
    for c in rows:
        esb = torch.exp2(s_base[c]) * w
        tables.egw[c] = (EG[c].reshape(2 * H, P)
                         * esb.unsqueeze(0)).to(torch.bfloat16)
        if tables.delta16 is not None:
            tables.delta16[c] = tables.delta[c].to(torch.float16)
        if tables.dall is not None:
            tables.dall[c] = tables.delta[c].sum(0)
