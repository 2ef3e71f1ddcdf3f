"""CODA: Consensus-Driven Active Model Selection.

Behavioral parity with the reference algorithm (coda/coda.py:171-346):
a Dirichlet confusion-matrix posterior per model, seeded from an
ensemble-consensus prior; P(model-is-best) via the diagonal-Beta grid
integral; expected-information-gain acquisition over unlabeled points;
greedy selection with seeded random tie-breaking.

MI355X-first differences from the reference implementation (none change
the math):
  - argmax classes and the disagreement prefilter mask are computed ONCE
    and cached (the reference recomputes both every step from the full
    (H,N,C) tensor: coda/coda.py:217,263).
  - the per-chunk hot path (hypothetical Beta updates -> Beta-grid P(best)
    -> EIG) is one fused op (`ops.eig_chunk`), which on GPU is a pair of
    hand-written gfx950 HIP kernels that never materialize the (R,H,P)
    pdf/cdf tensors (reference materializes ~6 of them:
    coda/coda.py:94-111).
  - the model axis may be sharded across ranks (`comm`); cross-model
    reductions become RCCL all-reduces (coda_amd/ops/sharded.py).
"""
from __future__ import annotations

import random
import struct

import torch
from sortedcontainers import SortedList

from ..base import ModelSelector
from ..parallel import Comm, get_comm
from .. import ops
from ..ops import sharded as shops
from ..util import DEBUG, _check

import os
DEBUG_VIZ = os.environ.get("CODA_AMD_DEBUG_VIZ") == "1"


class CODA(ModelSelector):
    def __init__(self, dataset,
                 prefilter_n: int = 0,
                 alpha: float = 0.9,
                 learning_rate: float = 0.01,
                 multiplier: float = 2.0,
                 disable_diag_prior: bool = False,
                 q: str = "eig",
                 comm: Comm = None,
                 chunk_size: int = 100,
                 num_points: int = 256,
                 pi_hat_precision: str = "auto",
                 eig_impl: str = "auto"):
        self.dataset = dataset
        self.device = dataset.preds.device
        self.comm = comm or get_comm()
        self.Hl, self.N, self.C = dataset.preds.shape  # local models
        self.H = getattr(dataset, "total_models", self.Hl)
        self.prefilter_n = prefilter_n
        self.disable_diag_prior = disable_diag_prior
        self.q = q
        self.chunk_size = chunk_size
        self.num_points = num_points
        # EIG implementation: 'fused' = the wave-per-row HIP kernel (GPU) /
        # eager (CPU); 'table' = the factored per-step curve tables + MFMA
        # GEMM (ops/table.py); 'pair' = the hit-sparse pair engine
        # (ops/pair.py - the full-pool no-prefilter fast path). 'auto'
        # resolves per device at eig time.
        self.eig_impl = eig_impl
        self._tables = None          # persistent v2 curve tables
        self._pairs_static = None    # v3 static hit structure (full pool)
        self._pair_row_of = None     # point id -> row in the static set
        self._active_mask = None     # (B_full,) bool device mask
        self._tables_dirty = set()   # class rows touched since last build
        self._posterior_version = 0  # bumped by add_label
        self._pbest_rows_cache = (-1, None)
        # hipGraph capture of the post-label update pipeline (Dirichlet row
        # update -> rank-1 pi_hat -> table row refresh -> posterior rows):
        # ~20 static-shape launches replayed as one graph. Auto on
        # single-device GPU; CODA_AMD_NO_GRAPH=1 disables.
        self._label_graph = None
        # hipGraph of the full-pool pair acquisition (static hit
        # structure + static table/posterior buffers -> the whole
        # get_next tensor pipeline incl. argmax/tie counting replays as
        # one graph; one host sync per step)
        self._acq_graph = None
        # (debug guards synchronize per op and cannot run under stream
        # capture - graph replay is a production-mode path)
        self._use_label_graph = (
            self.device.type == "cuda" and not self.comm.is_distributed
            and not DEBUG
            and os.environ.get("CODA_AMD_NO_GRAPH") != "1")

        # hyperparams (reference names: coda/coda.py:189-190)
        self.prior_strength = 1.0 - alpha
        self.update_strength = learning_rate

        preds = dataset.preds
        # one chunked fp32-upcast pass: cached argmax classes (Hl, N) +
        # consensus sum (storage may be fp32/bf16/fp8)
        self.classes, ens_sum = ops.init_model_stats(preds)

        # Sharded pair engine = "replicated Betas, sharded candidates":
        # every rank keeps the global (H, C) diagonal-Beta view + the
        # global argmax classes (gathered ONCE - they never change), so
        # tables/pbest/mixture are computed locally and bit-identically
        # on all ranks, EIG work is split by candidate, and the only
        # per-step wire is the (N,) pi_hat delta + a (B,) EIG gather.
        # This replaces the round-1 design that gathered (B, H, P) delta
        # curves per chunk (268 MB/chunk at the 8-GPU config;
        # VERDICT.md round-1 finding #1).
        self._replicated = (
            self.comm.is_distributed and self.H <= 1024
            and self.C <= 32767 and eig_impl in ("auto", "pair"))
        self.classes_global = None
        self._alpha_g = self._beta_g = None
        if self._replicated:
            # gather in int32 (gloo has no int16 collectives), store
            # int16 - at the 8-GPU 1024x1M config that's 2 GB resident
            cls32 = self.classes.to(torch.int32)
            gathered = self.comm.all_gather_cat(
                cls32, dim=0, sizes=self.comm.shard_sizes(self.H))
            order = self.comm.unshard_order(self.H).to(gathered.device)
            self.classes_global = gathered[order].to(torch.int16) \
                .contiguous()                               # (H, N)

        # pi_hat compute dtype: bf16 MFMA on GPU (one packed (N, H*C) GEMM,
        # ~16x the f32 matrix rate, f32 accumulation), fp32 elsewhere.
        if pi_hat_precision == "auto":
            pi_hat_precision = "bf16" if preds.is_cuda else "fp32"
        if preds.dtype == torch.float8_e4m3fn and pi_hat_precision == "fp32":
            pi_hat_precision = "bf16"  # fp8 storage has no fp32 bmm path
        self.pi_hat_precision = pi_hat_precision
        self._adjusted = None  # maintained incrementally after init

        # static class-major mirror of the prediction pool (H, C, N):
        # the per-label rank-1 pi_hat gather preds[h, :, cls_h] reads
        # one element per 64-B sector in row-major layout (at the
        # random-sector floor - 150 us headline, 1.2 ms at 10k models);
        # the mirror makes it a coalesced stream. Built only when HBM
        # has the pool's size + 24 GB headroom to spare (the 128 GB
        # fp8 million-point pool deliberately skips it).
        self._preds_t = None
        if preds.is_cuda and ops.hip_available() and preds.dim() == 3:
            try:
                free, _ = torch.cuda.mem_get_info(self.device)
                need = preds.numel() * preds.element_size()
                if free > need + 24 * (1 << 30):
                    self._preds_t = preds.permute(0, 2, 1).contiguous()
            except (RuntimeError, torch.OutOfMemoryError):
                self._preds_t = None

        # consensus prior: global mean over H (all-reduce site K1)
        self.comm.all_reduce_(ens_sum)
        pseudo = (ens_sum / self.H).argmax(-1)            # (N,) global pseudo-labels
        soft_conf = ops.confusion_prior(pseudo, preds)    # (Hl, C, C)
        self.dirichlets = ops.init_dirichlets(
            soft_conf, self.prior_strength, disable_diag_prior, multiplier)
        if self._replicated:
            # replicated global diagonal-Beta view (gathered once; kept
            # current locally from classes_global on every label)
            a_l, b_l = ops.dirichlet_to_beta(self.dirichlets)  # (Hl, C)
            sizes = self.comm.shard_sizes(self.H)
            order = self.comm.unshard_order(self.H).to(self.device)
            self._alpha_g = self.comm.all_gather_cat(
                a_l, dim=0, sizes=sizes)[order].contiguous()
            self._beta_g = self.comm.all_gather_cat(
                b_l, dim=0, sizes=sizes)[order].contiguous()
        self.update_pi_hat()

        # static disagreement mask (K12): "not all models agree".
        if self._replicated:
            ref = self.classes_global[0]
            self._disagreement = (self.classes_global
                                  != ref.unsqueeze(0)).any(dim=0)
        else:
            # Cross-shard: compare against global model 0's classes (it
            # lives on rank 0 under strided sharding), then OR-reduce.
            ref_row = self.classes[0].clone() if self.comm.rank == 0 \
                else torch.empty_like(self.classes[0])
            self.comm.broadcast_(ref_row, src=0)
            dis = (self.classes != ref_row.unsqueeze(0)).any(dim=0) \
                .to(torch.float32)
            self.comm.all_reduce_(dis)
            self._disagreement = dis > 0                  # (N,) bool
        # host-side copy for the per-step candidate filter (indexing a GPU
        # tensor point-by-point would be one device sync per point)
        self._disagreement_host = self._disagreement.cpu().tolist()
        # persistently maintained unlabeled-and-disagreeing candidates
        # (the mask is static; add_label removes one entry). SortedList
        # keeps the reference's ascending order with O(log N) removal -
        # a plain list.remove is an O(N) scan per label.
        self._active_candidates = SortedList(
            i for i in range(self.N) if self._disagreement_host[i])

        self.labeled_idxs, self.labels = [], []
        self.unlabeled_idxs = SortedList(range(self.N))
        self.q_vals = []
        self.stochastic = False
        self.step = 0

    @classmethod
    def from_args(cls, dataset, args, comm=None):
        return cls(dataset,
                   prefilter_n=args.prefilter_n,
                   alpha=args.alpha,
                   learning_rate=args.learning_rate,
                   multiplier=args.multiplier,
                   disable_diag_prior=args.no_diag_prior,
                   q=args.q,
                   comm=comm,
                   chunk_size=getattr(args, "chunk_size", 100),
                   pi_hat_precision=getattr(args, "pi_hat_precision",
                                            "auto"))

    # ------------------------------------------------------------------
    def _prefilter(self, idxs):
        """Drop points where every model agrees (no information), optionally
        subsample to prefilter_n (reference: coda/coda.py:215-224)."""
        if idxs is self.unlabeled_idxs:
            idxs = self._active_candidates  # incrementally maintained
        else:
            mask = self._disagreement_host
            idxs = [i for i in idxs if mask[i]]
        if self.prefilter_n and len(idxs) > self.prefilter_n:
            idxs = random.sample(idxs, self.prefilter_n)
            self.stochastic = True
        return idxs

    def update_pi_hat(self):
        """Confusion-adjusted class marginals (K5; all-reduce over H).

        Runs the FULL contraction (one packed bf16 MFMA GEMM on GPU).
        Called once at init and by checkpoint restore; per-label updates
        go through the exact rank-1 incremental path in add_label
        (ops.pi_hat_delta), which is O(H*N) instead of O(H*N*C^2).
        """
        if self.pi_hat_precision == "bf16":
            # N-chunked pack + bf16 MFMA GEMM: never holds the full
            # (N, H*C) operand, so million-point pools fit alongside the
            # prediction tensor in HBM
            adjusted = ops.pi_hat_partial_streamed(self.dirichlets,
                                                   self.dataset.preds)
        else:
            adjusted = ops.pi_hat_partial(self.dirichlets, self.dataset.preds)
        self.comm.all_reduce_(adjusted)
        self._adjusted = adjusted
        self._row_sums = adjusted.sum(dim=-1)
        self._refresh_pi_hat()

    def _refresh_pi_hat(self):
        """pi_hat (C,) from the maintained adjusted/row_sums WITHOUT
        materializing the (N, C) per-item normalization: the marginal is
        one GEMV, pi = adjusted^T @ (1/rowsum), then self-normalized
        (identical math to the reference's pi_hat_xi.sum(0) route,
        coda/coda.py:229-233). Per-item rows are built on demand
        (candidate gathers in eig_batched; the pi_hat_xi property)."""
        if (self._adjusted.is_cuda and self.C <= 2048
                and ops.hip_available()):
            pi = ops._ext.pi_marginal(self._adjusted, self._row_sums)
        else:
            inv = 1.0 / self._row_sums.clamp_min(1e-12)
            # row-vector @ matrix streams adjusted once; transposed mv is
            # 5x slower on ROCm (strided column reduction)
            pi = inv @ self._adjusted
        self.pi_hat = pi / pi.sum()
        self._pi_xi_cache = None

    @property
    def pi_hat_xi(self):
        """(N, C) per-item class posterior (materialized on demand)."""
        if self._pi_xi_cache is None:
            rs = self._row_sums.clamp_min(1e-12).unsqueeze(-1)
            self._pi_xi_cache = self._adjusted / rs
        return self._pi_xi_cache

    def _pi_xi_rows(self, ids):
        rs = self._row_sums[ids].clamp_min(1e-12).unsqueeze(-1)
        return self._adjusted[ids] / rs

    # ------------------------------------------------------------------
    def _pbest_rows_before(self):
        """(C, Hl) P(best | class row c) under the current posterior.

        Memoized on the posterior version: the rows computed by get_pbest
        right after a label are identical to the next acquisition's
        pbest_before (nothing moves the Dirichlets in between)."""
        ver, cached = self._pbest_rows_cache
        if ver == self._posterior_version and cached is not None:
            return cached
        alpha_cc, beta_cc = self._beta_view()
        a, b = alpha_cc.t().contiguous(), beta_cc.t().contiguous()
        if self.comm.is_distributed and not self._replicated:
            rows = shops.pbest_from_beta_sharded(a, b, self.comm,
                                                 self.num_points)
        else:
            # replicated mode computes the GLOBAL (C, H) rows locally
            rows = ops.pbest_from_beta(a, b, self.num_points)
        self._pbest_rows_cache = (self._posterior_version, rows)
        return rows

    def _beta_view(self):
        """(H?, C) diagonal Betas: global when replicated, else local."""
        if self._replicated:
            return self._alpha_g, self._beta_g
        return ops.dirichlet_to_beta(self.dirichlets)

    def _refresh_tables(self, alpha_cc, beta_cc, want_egw: bool):
        """Build or incrementally refresh the per-step curve tables."""
        from ..ops import table as tops
        if self._tables is None:
            self._tables = tops.table_precompute(
                alpha_cc, beta_cc, num_points=self.num_points)
        elif self._tables_dirty:
            tops.table_update_rows(self._tables, alpha_cc, beta_cc,
                                   sorted(self._tables_dirty))
        self._tables_dirty.clear()
        if want_egw and self._tables.egw is None:
            from ..ops import pair as pops
            self._tables = pops.attach_pair_tables(self._tables)
        return self._tables

    def _eig_pair(self, candidate_ids):
        """v3 hit-sparse EIG over the candidate set (ops/pair.py).

        Distributed: candidates are split round-robin by their position
        in the (shared, ascending) candidate list; each rank scores its
        slice against the replicated global tables, and the (B,) EIG
        values are all-gathered and re-interleaved - O(B) bytes per
        step on the wire.
        """
        from ..ops import pair as pops
        alpha_cc, beta_cc = self._beta_view()
        tables = self._refresh_tables(alpha_cc, beta_cc, want_egw=True)
        pbest_before = self._pbest_rows_before()            # (C, H)
        mixture0, H_before = ops.mixture_entropy(pbest_before, self.pi_hat)

        r, w = self.comm.rank, self.comm.world

        def build(ids_t):
            mine = ids_t[r::w] if self._replicated else ids_t
            cls_rows = self._global_classes(mine)
            wide = self.H > 1024  # beyond the vmask/fused-kernel regime
            return pops.build_pairs(
                cls_rows, mine, self.C, tile=(128 if wide else 0),
                with_vmask=not wide), cls_rows

        if candidate_ids is self._active_candidates:
            # full-pool acquisition: the hit structure is static (argmax
            # classes never change); labeled points are masked at gather
            if self._pairs_static is None:
                ids = torch.tensor(list(candidate_ids), device=self.device)
                self._pairs_static = build(ids)
                self._pair_row_of = {int(p): i
                                     for i, p in enumerate(ids.tolist())}
                self._active_mask = torch.ones(
                    ids.numel(), dtype=torch.bool, device=self.device)
                self._shard_perm = self._interleave_perm(ids.numel())
            ps, cls_rows = self._pairs_static
            mask, perm = self._active_mask, self._shard_perm
        else:
            cand = torch.tensor(list(candidate_ids), device=self.device)
            ps, cls_rows = build(cand)
            mask, perm = None, self._interleave_perm(cand.numel())
        if ps.cand_ids.numel():
            q = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                               self.pi_hat, mixture0, H_before,
                               self._adjusted, self._row_sums)
        else:  # more ranks than candidates
            q = torch.empty(0, device=self.device)
        if self._replicated:
            sizes = [len(range(rr, (mask.numel() if mask is not None
                                    else perm.numel()), w))
                     for rr in range(w)]
            q = self.comm.all_gather_cat(q, dim=0, sizes=sizes)[perm]
        return (q[mask] if mask is not None else q), candidate_ids

    def _interleave_perm(self, n: int) -> torch.Tensor:
        """Permutation mapping rank-concatenated round-robin slices back
        to list order (identity when not distributed)."""
        if not self._replicated:
            return torch.arange(n, device=self.device)
        w = self.comm.world
        perm = torch.empty(n, dtype=torch.long, device=self.device)
        off = 0
        for rr in range(w):
            cnt = len(range(rr, n, w))
            perm[rr:n:w] = torch.arange(off, off + cnt, device=self.device)
            off += cnt
        return perm

    def _hit_sparse(self) -> bool:
        """True when candidates hit few distinct classes (the pair
        engine's regime): estimated from a 1024-point sample once."""
        if not hasattr(self, "_est_distinct"):
            src = self.classes_global if self._replicated else self.classes
            n = src.shape[1]
            g = torch.Generator(device="cpu").manual_seed(0)
            idx = torch.randperm(n, generator=g)[:1024].to(src.device)
            sub = src[:, idx].long().sort(dim=0).values
            distinct = (sub[1:] != sub[:-1]).sum(0) + 1
            self._est_distinct = float(distinct.float().mean())
        return self._est_distinct <= max(self.C / 4.0, 32.0)

    def _global_classes(self, ids: torch.Tensor) -> torch.Tensor:
        """(B, H) int32 argmax classes of every GLOBAL model on the
        given points."""
        src = self.classes_global if self._replicated else self.classes
        return src[:, ids].t().to(torch.int32).contiguous()

    def eig_batched(self):
        """EIG for every candidate point (reference: coda/coda.py:235-281)."""
        candidate_ids = self._prefilter(self.unlabeled_idxs) \
            or self.unlabeled_idxs

        impl0 = self.eig_impl
        full_set = candidate_ids is self._active_candidates
        if impl0 == "auto":
            if self._replicated:
                impl0 = "pair"
            elif (self.device.type == "cuda"
                    and not self.comm.is_distributed and self.H <= 1024
                    and full_set and self._hit_sparse()):
                # full-pool acquisition: the static hit structure pays
                # for itself when candidates hit FEW distinct classes.
                # Prefiltered SUBSETS resample every step, where the v2
                # table chunks (tuned round 1: 1.33 ms/step at
                # prefilter 256) beat a per-step structure rebuild; and
                # when H >~ C the sparsity collapses (nearly every
                # class is hit by some model, K ~ B*C = the dense
                # problem) so v2 stays - measured 20.1 vs 4.0 ms/step
                # at H=4096, 15.2 vs ~5 at H=1024 full-pool.
                impl0 = "pair"
        if impl0 == "pair" and (self._replicated
                                or not self.comm.is_distributed):
            return self._eig_pair(candidate_ids)

        cand = torch.tensor(list(candidate_ids), device=self.device)

        pbest_before = self._pbest_rows_before()            # (C, Hl)
        alpha_cc, beta_cc = ops.dirichlet_to_beta(self.dirichlets)
        if self.comm.is_distributed:
            mixture0, H_before = shops.mixture_entropy_sharded(
                pbest_before, self.pi_hat, self.comm)
        else:
            mixture0, H_before = ops.mixture_entropy(pbest_before, self.pi_hat)

        impl = impl0
        if impl == "auto" or impl == "pair":
            impl = "table" if self.device.type == "cuda" else "fused"
        tables = s_base_all = None
        if impl == "table":
            from ..ops import table as tops
            tables = self._refresh_tables(alpha_cc, beta_cc,
                                          want_egw=False)
            if self.comm.is_distributed:
                s_base_all = tops.s_base_global(tables, self.comm)

        eig_chunks = []
        for s in range(0, cand.numel(), self.chunk_size):
            ids = cand[s:s + self.chunk_size]
            chunk_classes = self.classes[:, ids].t().contiguous()  # (B, Hl)
            pi_xi = self._pi_xi_rows(ids)                          # (B, C)
            if self.comm.is_distributed and tables is not None:
                from ..ops import table as tops
                eig = tops.eig_chunk_table_sharded(
                    tables, s_base_all, chunk_classes, pbest_before,
                    self.pi_hat, pi_xi, mixture0, H_before, self.comm,
                    hsizes=self.comm.shard_sizes(self.H))
            elif self.comm.is_distributed:
                eig = shops.eig_chunk_sharded(
                    alpha_cc, beta_cc, chunk_classes, pbest_before,
                    self.pi_hat, pi_xi, mixture0, H_before, self.comm,
                    num_points=self.num_points)
            elif tables is not None:
                from ..ops import table as tops
                eig = tops.eig_chunk_table(
                    tables, chunk_classes, pbest_before, self.pi_hat,
                    pi_xi, mixture0, H_before)
            else:
                eig = ops.eig_chunk(
                    alpha_cc, beta_cc, chunk_classes, pbest_before,
                    self.pi_hat, pi_xi, mixture0, H_before,
                    num_points=self.num_points)
            eig_chunks.append(eig)

        return torch.cat(eig_chunks), candidate_ids

    # -- hipGraph full-pool acquisition --------------------------------
    def _acq_body(self):
        """The whole get_next tensor pipeline on the STATIC buffers:
        mixture entropy from the label-graph's posterior rows/pi_hat,
        the three pair kernels, labeled-candidate masking, argmax and
        tie counting. Captured once; identical math to _eig_pair."""
        t = self._tables
        ps, _ = self._pairs_static
        rows, pi = self._g_rows, self._g_pi
        mixture0, H0 = ops.mixture_entropy(rows, pi)
        A16 = ops._ext.pair_dsum_es(t.delta16, t.dall, ps.pair_c,
                                    ps.pair_neg, ps.seg_off, ps.seg_h)
        grp = (ps.grp_off if ps.grp_off is not None
               else torch.empty(0, dtype=torch.int32, device=A16.device))
        h_after = ops._ext.pair_gemm_entropy(
            A16, t.egw, ps.vmask, ps.pair_c, pi, rows,
            mixture0.contiguous(), ps.tile, 0, grp)
        # H_before enters as an in-graph tensor op (the kernel's scalar
        # argument would be frozen at capture value)
        h_base = h_after.index_select(0, ps.base_pos).contiguous()
        q0 = ops._ext.pair_eig_finalize(
            h_after, h_base, ps.pair_c, ps.cand_off,
            ps.cand_ck, ps.cand_ids, self._adjusted, self._row_sums,
            0.0)
        # fused epilogue: qbuf = active ? H0 + q0 : -inf; _acq_out =
        # [masked max, first argmax, isclose tie count].  Replaces the
        # ~10-launch torch chain (where/max/isclose/sum/copies,
        # ~120 us/step at B=50k); identical semantics - when the tie
        # count is 1 the argmax is unique, and ties go through the
        # seeded host random.choice either way (_acq_result).
        ops._ext.acq_select(q0.contiguous(), H0.reshape(1).contiguous(),
                            self._active_mask, self._acq_qbuf,
                            self._acq_out, self._acq_ties)

    def _acq_result(self):
        self._acq_out_host.copy_(self._acq_out)
        bv, bi, nt = self._acq_out_host.tolist()
        bi, nt = int(bi), int(nt)
        if bi < 0:
            raise RuntimeError("acquisition ran with no active "
                               "candidates (pool exhausted?)")
        if nt > 1:
            # same tie semantics as the eager path: active candidates
            # ascend by point id in both orderings.  The fused epilogue
            # collected (position, q value) pairs already (unordered
            # slots); sorting restores the ascending order the seeded
            # random.choice tie-break expects.
            if nt < self._acq_ties.numel():
                th = self._acq_ties_host[:1 + nt]
                th.copy_(self._acq_ties[:1 + nt])
                packed = sorted(v & 0xFFFFFFFFFFFFFFFF
                                for v in th[1:].tolist())
                pk = random.choice(packed)
                pos = pk >> 32
                qv = struct.unpack(
                    "<f", struct.pack("<I", pk & 0xFFFFFFFF))[0]
            else:  # > buffer capacity: rescan (never seen in practice)
                q = self._acq_qbuf
                ties = (torch.isclose(q, q.max(), rtol=1e-8)
                        & self._active_mask)
                pos = random.choice(
                    torch.nonzero(ties, as_tuple=True)[0].tolist())
                qv = float(q[pos])
            self.stochastic = True
            return self._pairs_ids_host[pos], qv
        return self._pairs_ids_host[bi], bv

    def _init_acq_buffers(self):
        if getattr(self, "_acq_out", None) is None:
            ps, _ = self._pairs_static
            self._pairs_ids_host = ps.cand_ids.cpu().tolist()
            self._acq_out = torch.zeros(3, dtype=torch.float64,
                                        device=self.device)
            self._acq_qbuf = torch.empty(ps.cand_ids.numel(),
                                         device=self.device)
            # fused-epilogue tie buffer: [0]=slot counter, [1:] packs
            # (position << 32) | q-value-bits per tie (host-sorted back
            # to ascending position - one D2H fetch serves both the
            # seeded tie-break AND its q value); overflow falls back to
            # the full-scan path.  Host mirrors are PINNED so the
            # per-step result fetches take the fast DMA path.
            self._acq_ties = torch.zeros(8192, dtype=torch.int64,
                                         device=self.device)
            self._acq_out_host = torch.empty(
                3, dtype=torch.float64, pin_memory=True)
            self._acq_ties_host = torch.empty(
                8192, dtype=torch.int64, pin_memory=True)

    def _acq_eligible(self) -> bool:
        return (self.q == "eig" and self._pairs_static is not None
                and self._pairs_static[0].vmask is not None
                and not DEBUG_VIZ
                and not (self.prefilter_n
                         and len(self._active_candidates)
                         > self.prefilter_n))

    def _graphed_acquire(self):
        if self._acq_graph is None:
            self._init_acq_buffers()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                self._acq_body()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            first = self._acq_result()  # before capture rebinds _acq_q
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._acq_body()
            self._acq_graph = g
            return first
        self._acq_graph.replay()
        return self._acq_result()

    # ------------------------------------------------------------------
    def get_next_item_to_label(self):
        if getattr(self, "_acq_fresh", False) and self._acq_eligible():
            # the merged label+acquire graph already computed this
            # step's EIG/argmax during add_label
            self._acq_fresh = False
            if getattr(self, "_acq_saved", None) is not None:
                res, self._acq_saved = self._acq_saved, None
                return res
            return self._acq_result()
        if (self.q == "eig" and self._use_label_graph
                and self._label_graph is not None
                and self._acq_eligible()):
            return self._graphed_acquire()
        if self.q == "eig":
            q_vals, cand = self.eig_batched()
        elif self.q == "iid":
            cand = self._prefilter(self.unlabeled_idxs) or self.unlabeled_idxs
            q_vals = torch.full((len(cand),), 1.0 / len(cand),
                                device=self.device)
        elif self.q == "uncertainty":
            cand = self._prefilter(self.unlabeled_idxs) or self.unlabeled_idxs
            ens_sum = self.dataset.preds.sum(dim=0)
            self.comm.all_reduce_(ens_sum)
            ent = ops.entropy_acquisition(ens_sum / self.H)
            q_vals = ent[torch.tensor(list(cand), device=self.device)]
        else:
            raise NotImplementedError(self.q)

        if DEBUG_VIZ:  # per-step EIG bar chart (reference coda/coda.py:299-303)
            from ..util import plot_bar
            from .. import tracking
            try:
                tracking.log_image(plot_bar(q_vals, title="EIG"),
                                   key="EIG", step=self.step)
            except RuntimeError:
                pass  # no active tracking run

        # greedy with seeded random tie-breaking (coda/coda.py:306-313);
        # max/argmax/tie-count fetched in ONE device sync. The index and
        # count travel as a separate int64 pair (a float32 round-trip
        # corrupts indices beyond 2^24, i.e. pools over ~16.7M points).
        best_val, best_idx = q_vals.max(0)
        n_ties = torch.isclose(q_vals, best_val, rtol=1e-8).sum()
        bv, bi, nt = torch.stack(
            [best_val.double(), best_idx.double(),
             n_ties.double()]).cpu().tolist()
        bi, nt = int(bi), int(nt)
        if nt > 1:
            ties = torch.isclose(q_vals, best_val, rtol=1e-8)
            idx_local = random.choice(
                torch.nonzero(ties, as_tuple=True)[0].tolist())
            self.stochastic = True
            return cand[idx_local], float(q_vals[idx_local])
        return cand[int(bi)], bv

    # -- hipGraph label-update pipeline --------------------------------
    def _label_update_body(self):
        """The tensor part of add_label as static-shape ops on the static
        input buffers self._g_idx / self._g_y (both (1,) int64). Writes
        tables in place and leaves the updated posterior rows in
        self._g_rows. Captured once into a hipGraph; identical math to
        the eager path (it IS the eager path, replayed)."""
        lr = self.update_strength
        idx_t, y_t = self._g_idx, self._g_y
        col = self.classes.index_select(1, idx_t).squeeze(1)     # (Hl,)
        if self.dirichlets.is_cuda and ops.hip_available():
            # H-thread scatter: torch's one_hot + dim-1 index_add_ on the
            # (H,C,C) posterior costs ~914 us for H real nonzeros
            ops._ext.dirichlet_add(self.dirichlets, y_t, col, float(lr))
        else:
            onehot = torch.nn.functional.one_hot(col, self.C).to(
                self.dirichlets.dtype)
            self.dirichlets.index_add_(1, y_t, (lr * onehot).unsqueeze(1))
        delta = ops.pi_hat_delta(self.dataset.preds, col,
                                 preds_t=self._preds_t) * lr     # (N,)
        if self._adjusted.is_cuda and ops.hip_available():
            # fused column update (torch index_add_ over dim 1 with one
            # index runs ~90x slower than this elementwise pass)
            ops._ext.col_add(self._adjusted, self._row_sums, y_t, delta)
        else:
            self._adjusted.index_add_(1, y_t, delta.unsqueeze(1))
            self._row_sums += delta
        if self.C <= 2048 and ops.hip_available():
            pi = ops._ext.pi_marginal(self._adjusted, self._row_sums)
        else:
            pi = (1.0 / self._row_sums.clamp_min(1e-12)) @ self._adjusted
        self._g_pi.copy_(pi / pi.sum())
        # refresh the labeled class's table row (v2 tables)
        if self._tables is not None:
            t = self._tables
            if (t.eg16 is not None and t.egw is not None
                    and t.delta16 is not None and t.dall is not None
                    and ops.hip_available()):
                # one fused commit (trc_* kernels in pbest.hip): the
                # torch chain below was ~14 in-graph launches incl. two
                # 32-thread strided reductions (10-24 us each)
                a_col, b_col = ops._ext.table_commit_row(
                    self.dirichlets, y_t, t.EG, t.delta, t.s_base,
                    t.weights, t.eg16, t.egw, t.delta16, t.dall, 1.0)
            else:
                row = self.dirichlets.index_select(1, y_t) \
                    .squeeze(1)                            # (Hl, C)
                a_col = row.gather(
                    1, y_t.view(1, 1).expand(self.Hl, 1)) \
                    .squeeze(1).contiguous()
                b_col = (row.sum(1) - a_col).contiguous()
                eg, lc = ops._ext.beta_row_tables(a_col, b_col, 1.0)
                t.EG.index_copy_(0, y_t, eg.unsqueeze(0))
                if t.eg16 is not None:
                    t.eg16.index_copy_(
                        0, y_t, eg.reshape(1, 2 * self.Hl, -1)
                        .to(torch.bfloat16))
                t.delta.index_copy_(0, y_t,
                                    (lc[:, 1] - lc[:, 0]).unsqueeze(0))
                t.s_base.index_copy_(0, y_t, lc[:, 0].sum(0).unsqueeze(0))
                if t.dall is not None:
                    t.dall.index_copy_(
                        0, y_t, t.delta.index_select(0, y_t).sum(1))
                if t.egw is not None:
                    esb = torch.exp2(t.s_base.index_select(0, y_t)) \
                        * t.weights                         # (1, P)
                    t.egw.index_copy_(
                        0, y_t, (eg.reshape(1, 2 * self.Hl, -1)
                                 * esb.unsqueeze(1)).to(torch.bfloat16))
                    t.delta16.index_copy_(
                        0, y_t, t.delta.index_select(0, y_t)
                        .to(torch.float16))
        # posterior rows: add_label moves only Dirichlet row y, so only
        # class y's Beta column - hence only pbest row y - changes.
        # _g_rows was seeded with the full (C, Hl) rows at graph init;
        # each label refreshes one row (was a full C-row recompute,
        # 103 us/step at C=1000).
        rows_y = ops.pbest_from_beta(a_col.unsqueeze(0),
                                     b_col.unsqueeze(0),
                                     self.num_points)        # (1, Hl)
        self._g_rows.index_copy_(0, y_t, rows_y)

    def _graphed_add_label(self, idx: int, true_class: int):
        if self._label_graph is None:
            self._g_idx = torch.zeros(1, dtype=torch.long,
                                      device=self.device)
            self._g_y = torch.zeros(1, dtype=torch.long, device=self.device)
            self._g_pi = torch.empty_like(self.pi_hat)
            # seeded with the PRE-label rows; the graph body refreshes
            # only the labeled class's row per replay
            self._g_rows = self._pbest_rows_before().clone().contiguous()
            self._g_idx.fill_(idx)
            self._g_y.fill_(true_class)
            # When the full-pool pair acquisition is active, the NEXT
            # step's acquisition is captured INTO the label graph: one
            # replay per step computes posterior update + next EIG +
            # argmax (the caller's get_next just reads the result).
            self._merged_acq = self._acq_eligible()
            if self._merged_acq:
                self._init_acq_buffers()

            def body():
                self._label_update_body()
                if self._merged_acq:
                    self._acq_body()

            # warmup executes the body for real (THIS label's update) on
            # a side stream, as stream capture requires; the capture pass
            # then only RECORDS the ops (no execution, no state change),
            # so the first label must not additionally replay.
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                body()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            if self._merged_acq:
                # the warmup's acquisition result serves the next
                # get_next (capture below records without executing)
                self._acq_saved = self._acq_result()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                body()
            self._label_graph = g
            self.pi_hat = self._g_pi
            self._pbest_rows_cache = (self._posterior_version + 1,
                                      self._g_rows)
            self._acq_fresh = self._merged_acq
            return
        self._g_idx.fill_(idx)
        self._g_y.fill_(true_class)
        self._label_graph.replay()
        self.pi_hat = self._g_pi
        self._pbest_rows_cache = (self._posterior_version + 1,
                                  self._g_rows)
        self._acq_fresh = self._merged_acq

    def add_label(self, idx, true_class, selection_prob):
        """Posterior update (K13) + incremental pi_hat refresh.

        Reference semantics (coda/coda.py:315-323); the pi_hat refresh is
        the exact rank-1 increment (only Dirichlet row `true_class`
        moved), all-reduced over shards. On single-device GPU the whole
        tensor pipeline replays as one hipGraph."""
        idx = int(idx)
        if (self._use_label_graph and self._tables is not None
                and not self._tables_dirty):
            # the candidate mask update must be enqueued BEFORE the
            # replay: the merged graph's acquisition half reads it
            self._deactivate(idx)
            self._graphed_add_label(idx, int(true_class))
            self._posterior_version += 1
            self._pi_xi_cache = None
            self.labeled_idxs.append(idx)
            self.labels.append(int(true_class))
            self.q_vals.append(selection_prob)
            self.unlabeled_idxs.remove(idx)
            return
        if self.dirichlets.is_cuda and ops.hip_available():
            # H-thread scatter kernel; torch's one_hot + dim-1 index_add_
            # on (H, C, C) costs ~915 us (matters on the eager path the
            # distributed ranks take - no hipGraph under collectives)
            y_t = torch.tensor([int(true_class)], dtype=torch.long,
                               device=self.device)
            ops._ext.dirichlet_add(self.dirichlets, y_t,
                                   self.classes[:, idx].contiguous(),
                                   float(self.update_strength))
        else:
            onehot = torch.nn.functional.one_hot(
                self.classes[:, idx], self.C).to(self.dirichlets.dtype)
            self.dirichlets[:, int(true_class)] += \
                self.update_strength * onehot
        if self._replicated:
            # keep the replicated global Beta view current - a local
            # elementwise update from the (gathered-once) global classes,
            # identical on every rank, no collective
            hit = self.classes_global[:, idx].long() == int(true_class)
            self._alpha_g[:, int(true_class)] += \
                self.update_strength * hit.float()
            self._beta_g[:, int(true_class)] += \
                self.update_strength * (~hit).float()
        self._tables_dirty.add(int(true_class))
        self._posterior_version += 1
        delta = ops.pi_hat_delta(self.dataset.preds, self.classes[:, idx],
                                 preds_t=self._preds_t)
        self.comm.all_reduce_(delta)
        self._adjusted[:, int(true_class)] += self.update_strength * delta
        self._row_sums += self.update_strength * delta
        self._refresh_pi_hat()
        # incremental posterior rows on the EAGER path too (the
        # distributed ranks and graph-disabled runs): only class y's
        # Beta column moved, so refresh one row instead of the full
        # (C, H) recompute (~300 us/step at C=1000) the version-keyed
        # cache would otherwise trigger.  Valid whenever this process
        # holds the full model axis (single device or replicated mode);
        # the H-sharded v2 path keeps the sharded recompute.
        ver, cached = self._pbest_rows_cache
        if (cached is not None and ver == self._posterior_version - 1
                and (self._replicated or not self.comm.is_distributed)):
            y = int(true_class)
            if self._replicated:
                a_col = self._alpha_g[:, y].contiguous()
                b_col = self._beta_g[:, y].contiguous()
            else:
                row = self.dirichlets[:, y, :]
                a_col = row[:, y].contiguous()
                b_col = (row.sum(-1) - a_col).contiguous()
            rows = cached.clone()
            rows[y] = ops.pbest_from_beta(
                a_col.unsqueeze(0), b_col.unsqueeze(0),
                self.num_points)[0]
            self._pbest_rows_cache = (self._posterior_version, rows)
        self.labeled_idxs.append(idx)
        self.labels.append(int(true_class))
        self.q_vals.append(selection_prob)
        self.unlabeled_idxs.remove(idx)
        self._deactivate(idx)

    def _deactivate(self, idx: int):
        """Remove a point from the candidate pool (labeled or skipped)."""
        try:
            self._active_candidates.remove(idx)
        except ValueError:
            pass  # point was not a disagreeing candidate
        if self._pair_row_of is not None:
            row = self._pair_row_of.get(idx)
            if row is not None:
                self._active_mask[row] = False

    def skip(self, idx):
        """Drop a point without labeling it (the demo's "I don't know"
        path, reference demo/app.py:188-189)."""
        idx = int(idx)
        if idx in self.unlabeled_idxs:
            self.unlabeled_idxs.remove(idx)
        self._deactivate(idx)
        self._acq_fresh = False  # precomputed EIG predates the skip

    # ------------------------------------------------------------------
    def get_pbest(self):
        """Marginal P(best) over models (K15): (H,) in GLOBAL model order."""
        rows = self._pbest_rows_before()     # (C, Hl) / (C, H) replicated
        # ops.mixture_entropy's column-sum kernel; the torch reduction
        # over C launches 32 threads (~22 us at C=1000)
        marg_local = ops.mixture_entropy(rows, self.pi_hat)[0]
        if self.comm.is_distributed and not self._replicated:
            gathered = self.comm.all_gather_cat(marg_local, dim=0)
            pbest = gathered[self.comm.unshard_order(self.H).to(gathered.device)]
        else:
            pbest = marg_local
        if DEBUG:
            _check(pbest, "Pbest")
        return pbest

    def get_best_model_prediction(self):
        pbest = self.get_pbest()
        if DEBUG_VIZ:  # reference coda/coda.py:337-341
            from ..util import plot_bar
            from .. import tracking
            try:
                tracking.log_image(plot_bar(pbest, title="PBest"),
                                   key="PBest", step=self.step)
            except RuntimeError:
                pass
        self.step += 1
        return torch.argmax(pbest)
