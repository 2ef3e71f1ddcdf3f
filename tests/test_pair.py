"""v3 pair-factored EIG engine (coda_amd/ops/pair.py), CPU.

The pair engine must reproduce the dense table engine (ops/table.py) and
the fused eager formulation (ops/reference.py) exactly up to fp32
reduction order: same EIG values, same selections, same trajectories.
Reference semantics: coda/coda.py:235-281 with --prefilter-n 0.
"""
from __future__ import annotations

import random

import pytest
import torch

from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.options import LOSS_FNS
from coda_amd.ops import pair as pops
from coda_amd.ops import table as tops
from coda_amd.ops import reference as R


def _random_problem(H=10, N=120, C=7, seed=0):
    g = torch.Generator().manual_seed(seed)
    preds = torch.softmax(torch.randn(H, N, C, generator=g) * 2, dim=-1)
    cls = preds.argmax(-1)                     # (H, N)
    dirichlets = torch.rand(H, C, C, generator=g) * 3 + 0.5
    pi_hat = torch.rand(C, generator=g)
    pi_hat /= pi_hat.sum()
    adjusted = torch.rand(N, C, generator=g) + 0.1
    row_sums = adjusted.sum(-1)
    return preds, cls, dirichlets, pi_hat, adjusted, row_sums


class TestPairStructure:
    def test_covers_every_hit_exactly_once(self):
        """Every (candidate, class) hit maps through the CSR to exactly
        one representative pair whose (class, hit-set) matches the
        brute-force set (pairs with identical sets are deduplicated)."""
        _, cls, *_ = _random_problem(H=9, N=50, C=6, seed=1)
        ids = torch.arange(37)
        cls_rows = cls[:, ids].t().contiguous()
        ps = pops.build_pairs(cls_rows, ids, 6)

        def pair_set(k):
            hs = set(ps.seg_h[int(ps.seg_off[k]):
                              int(ps.seg_off[k + 1])].tolist())
            if int(ps.pair_neg[k]):
                hs = set(range(9)) - hs
            return hs

        seen = {}
        for b in range(37):
            for s in range(int(ps.cand_off[b]), int(ps.cand_off[b + 1])):
                k = int(ps.cand_pairs[s])
                c = int(ps.pair_c[k])
                assert (b, c) not in seen, "duplicate CSR entry"
                seen[(b, c)] = pair_set(k)
        for b in range(37):
            for c in range(6):
                expect = {h for h in range(9) if int(cls_rows[b, h]) == c}
                if expect:
                    assert seen[(b, c)] == expect, (b, c)
                else:
                    assert (b, c) not in seen
        # dedupe really collapses: fewer evaluated pairs than hits
        n_hits = len(seen)
        assert ps.n_real <= n_hits
        # base/pad pairs have empty segments and no complement flag
        for k in range(ps.K):
            if int(ps.pair_b[k]) < 0:
                assert int(ps.seg_off[k + 1]) == int(ps.seg_off[k])
                assert int(ps.pair_neg[k]) == 0

    def test_tiles_are_class_uniform_and_base_marked(self):
        _, cls, *_ = _random_problem(H=12, N=64, C=5, seed=2)
        ids = torch.arange(64)
        ps = pops.build_pairs(cls[:, ids].t().contiguous(), ids, 5)
        assert ps.K % ps.tile == 0
        pc = ps.pair_c.view(-1, ps.tile)
        assert (pc == pc[:, :1]).all(), "tile straddles classes"
        # base pair of every class exists, has empty segment, b = -1
        for c in range(5):
            k = int(ps.base_pos[c])
            assert int(ps.pair_c[k]) == c
            assert int(ps.pair_b[k]) == -1
            assert int(ps.seg_off[k + 1]) == int(ps.seg_off[k])


class TestPairEig:
    @pytest.mark.parametrize("H,N,C", [(10, 120, 7), (3, 60, 2),
                                       (16, 200, 126)])
    def test_matches_table_engine(self, H, N, C):
        (preds, cls, dirichlets, pi_hat, adjusted,
         row_sums) = _random_problem(H, N, C, seed=H + C)
        alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
        tables = tops.table_precompute(alpha_cc, beta_cc)
        pbest_before = R.pbest_from_beta(alpha_cc.t().contiguous(),
                                         beta_cc.t().contiguous())
        mixture0, H_before = R.mixture_entropy(pbest_before, pi_hat)

        ids = torch.arange(N)
        cls_rows = cls[:, ids].t().contiguous()
        ps = pops.build_pairs(cls_rows, ids, C)
        eig_n = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                               pi_hat, mixture0, H_before, adjusted,
                               row_sums)

        pi_xi = adjusted / row_sums.clamp_min(1e-12).unsqueeze(-1)
        eig_dense = tops.eig_chunk_table(tables, cls_rows, pbest_before,
                                         pi_hat, pi_xi, mixture0,
                                         H_before)
        torch.testing.assert_close(eig_n, eig_dense, rtol=2e-4,
                                   atol=1e-6)

    def test_trajectory_equals_fused_and_table(self):
        preds, labels = make_synthetic_task(H=12, N=400, C=9, seed=3)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        oracle = Oracle(ds, LOSS_FNS["acc"])

        def run(impl, prefilter):
            random.seed(0)
            torch.manual_seed(0)
            sel = CODA(ds, eig_impl=impl, prefilter_n=prefilter,
                       chunk_size=64)
            traj = []
            for _ in range(6):
                i, q = sel.get_next_item_to_label()
                sel.add_label(i, oracle(int(i)), q)
                traj.append((int(i),
                             int(sel.get_best_model_prediction())))
            return traj, sel.get_pbest()

        for prefilter in (0, 32):
            t_f, p_f = run("fused", prefilter)
            t_t, _ = run("table", prefilter)
            t_p, p_p = run("pair", prefilter)
            assert t_f == t_t == t_p
            torch.testing.assert_close(p_f, p_p, rtol=1e-4, atol=1e-6)

    def test_static_structure_reused_and_masked(self):
        preds, labels = make_synthetic_task(H=8, N=150, C=5, seed=4)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0)
        torch.manual_seed(0)
        sel = CODA(ds, eig_impl="pair")
        i0, q0 = sel.get_next_item_to_label()
        assert sel._pairs_static is not None
        ps0 = sel._pairs_static[0]
        sel.add_label(i0, oracle(int(i0)), q0)
        i1, _ = sel.get_next_item_to_label()
        assert sel._pairs_static[0] is ps0, "structure rebuilt"
        assert int(i1) != int(i0), "labeled point re-selected"
        row = sel._pair_row_of[int(i0)]
        assert not bool(sel._active_mask[row])

    def test_skip_removes_candidate(self):
        preds, labels = make_synthetic_task(H=8, N=150, C=5, seed=5)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        random.seed(0)
        torch.manual_seed(0)
        sel = CODA(ds, eig_impl="pair")
        i0, _ = sel.get_next_item_to_label()
        sel.skip(i0)
        i1, _ = sel.get_next_item_to_label()
        assert int(i1) != int(i0)


class TestDedupeProperties:
    @pytest.mark.parametrize("H,N,C,seed", [(9, 200, 6, 0), (16, 300, 3, 1),
                                            (5, 100, 50, 2), (32, 150, 8, 3)])
    def test_dedupe_equals_no_dedupe(self, H, N, C, seed):
        """EIG through the deduplicated structure == through the
        1:1 structure, for random shapes (guards the set-hash grouping
        and the CSR remap)."""
        (preds, cls, dirichlets, pi_hat, adjusted,
         row_sums) = _random_problem(H, N, C, seed=seed)
        alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
        tables = tops.table_precompute(alpha_cc, beta_cc)
        pbest_before = R.pbest_from_beta(alpha_cc.t().contiguous(),
                                         beta_cc.t().contiguous())
        mixture0, H_before = R.mixture_entropy(pbest_before, pi_hat)
        ids = torch.arange(N)
        cls_rows = cls[:, ids].t().contiguous()
        out = {}
        for dd in (True, False):
            ps = pops.build_pairs(cls_rows, ids, C, dedupe=dd)
            out[dd] = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                                     pi_hat, mixture0, H_before,
                                     adjusted, row_sums)
        assert out[True].shape == out[False].shape
        torch.testing.assert_close(out[True], out[False], rtol=1e-5,
                                   atol=1e-7)

    def test_strided_candidate_slice_matches_full(self):
        """Slice structures (the sharded per-rank view, incl. strided
        cand_ids) reproduce the full-pool EIG exactly on CPU."""
        (preds, cls, dirichlets, pi_hat, adjusted,
         row_sums) = _random_problem(11, 180, 7, seed=9)
        alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
        tables = tops.table_precompute(alpha_cc, beta_cc)
        pbest_before = R.pbest_from_beta(alpha_cc.t().contiguous(),
                                         beta_cc.t().contiguous())
        mixture0, H_before = R.mixture_entropy(pbest_before, pi_hat)
        ids = torch.arange(180)
        cls_all = cls[:, ids].t().contiguous()
        ps = pops.build_pairs(cls_all, ids, 7)
        q_full = pops.eig_pairs(tables, ps, cls_all, pbest_before,
                                pi_hat, mixture0, H_before, adjusted,
                                row_sums)
        for r in range(3):
            mine = ids[r::3]             # non-contiguous slice
            cls_r = cls[:, mine].t().contiguous()
            ps_r = pops.build_pairs(cls_r, mine, 7)
            assert ps_r.cand_ids.is_contiguous()
            q_r = pops.eig_pairs(tables, ps_r, cls_r, pbest_before,
                                 pi_hat, mixture0, H_before, adjusted,
                                 row_sums)
            torch.testing.assert_close(q_r, q_full[r::3], rtol=1e-6,
                                       atol=1e-8)
