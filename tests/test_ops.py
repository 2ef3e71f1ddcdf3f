"""Golden tests for the compute core vs fp64 NumPy/SciPy references.

Each op in coda_amd/ops/reference.py is checked against an independent
fp64 implementation (scipy.stats.beta for the P(best) integral, dense
broadcasts for the structured ops). The HIP kernels are checked against
these same eager ops in tests/test_gpu.py.
"""
import numpy as np
import scipy.stats
import torch

from coda_amd import ops
from coda_amd.ops import reference as R


def _rand_betas(rows=7, H=5, seed=0, lo=0.5, hi=60.0):
    g = torch.Generator().manual_seed(seed)
    a = torch.rand(rows, H, generator=g) * (hi - lo) + lo
    b = torch.rand(rows, H, generator=g) * (hi - lo) + lo
    return a, b


def _pbest_fp64(a, b, P=256):
    """fp64 NumPy golden for the Beta-grid P(best) op."""
    a = a.double().numpy()
    b = b.double().numpy()
    x = np.linspace(1e-6, 1 - 1e-6, P)
    out = np.zeros_like(a)
    for r in range(a.shape[0]):
        pdf = np.stack([scipy.stats.beta.pdf(x, a[r, h], b[r, h])
                        for h in range(a.shape[1])])
        # trapezoid cumulative
        cdf = np.zeros_like(pdf)
        cdf[:, 1:] = np.cumsum(0.5 * (pdf[:, 1:] + pdf[:, :-1]) *
                               np.diff(x), axis=1)
        logc = np.log(np.clip(cdf, 1e-30, None))
        prod_excl = np.exp(np.clip(logc.sum(0) - logc, -80, 80))
        integ = pdf * prod_excl
        p = np.trapezoid(integ, x, axis=1)
        out[r] = p / max(p.sum(), 1e-30)
    return out


class TestPbest:
    def test_matches_fp64_scipy(self):
        a, b = _rand_betas()
        got = ops.pbest_from_beta(a, b).numpy()
        want = _pbest_fp64(a, b)
        np.testing.assert_allclose(got, want, rtol=2e-4, atol=2e-6)

    def test_rows_sum_to_one(self):
        a, b = _rand_betas(rows=20, H=16, seed=3)
        p = ops.pbest_from_beta(a, b)
        assert torch.isfinite(p).all()
        np.testing.assert_allclose(p.sum(-1).numpy(), 1.0, atol=1e-4)

    def test_two_model_closed_form(self):
        """H=2: P(X > Y) for X~Beta(a1,b1), Y~Beta(a2,b2) via scipy dblquad
        of the exact density - grid-free ground truth."""
        a = torch.tensor([[8.0, 4.0]])
        b = torch.tensor([[4.0, 8.0]])
        got = ops.pbest_from_beta(a, b)[0]

        from scipy.integrate import quad
        def integrand(x):
            return (scipy.stats.beta.pdf(x, 8, 4) *
                    scipy.stats.beta.cdf(x, 4, 8))
        p1, _ = quad(integrand, 0, 1)
        # the grid integral is approximate; 256 points gives ~1e-3
        assert abs(float(got[0]) - p1) < 5e-3
        assert abs(float(got[0] + got[1]) - 1.0) < 1e-5

    def test_dominant_model_wins(self):
        a = torch.tensor([[50.0, 10.0, 10.0]])
        b = torch.tensor([[10.0, 50.0, 50.0]])
        p = ops.pbest_from_beta(a, b)[0]
        assert p[0] > 0.99

    def test_large_params_finite(self):
        """Concentrated Betas (large counts) must stay finite (the log-space
        +-80 clamp, reference coda/coda.py:104-107)."""
        a, b = _rand_betas(rows=4, H=8, seed=5, lo=100.0, hi=5000.0)
        p = ops.pbest_from_beta(a, b)
        assert torch.isfinite(p).all()
        np.testing.assert_allclose(p.sum(-1).numpy(), 1.0, atol=1e-3)


class TestPriorOps:
    def test_confusion_prior_vs_onehot_einsum(self):
        g = torch.Generator().manual_seed(1)
        H, N, C = 4, 50, 6
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1)
        labels = torch.randint(0, C, (N,), generator=g)
        got = ops.confusion_prior(labels, preds)
        onehot = torch.nn.functional.one_hot(labels, C).float()
        conf = torch.einsum("nc,hnj->hcj", onehot, preds)
        want = conf / conf.sum(-1, keepdim=True).clamp_min(1e-6)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)

    def test_init_dirichlets_diag(self):
        sc = torch.rand(3, 4, 4)
        d = ops.init_dirichlets(sc, 0.1, False, 2.0)
        base = torch.full((4, 4), 1 / 3.0)
        base.fill_diagonal_(1.0)
        torch.testing.assert_close(d, 2.0 * (base + 0.1 * sc))

    def test_init_dirichlets_uniform(self):
        sc = torch.rand(3, 4, 4)
        d = ops.init_dirichlets(sc, 0.1, True, 1.0)
        torch.testing.assert_close(d, 0.5 + 0.1 * sc)

    def test_dirichlet_to_beta(self):
        d = torch.rand(5, 4, 4) + 0.5
        a, b = ops.dirichlet_to_beta(d)
        for h in range(5):
            for c in range(4):
                assert abs(a[h, c] - d[h, c, c]) < 1e-6
                assert abs(b[h, c] - (d[h, c].sum() - d[h, c, c])) < 1e-5

    def test_pi_hat_vs_einsum(self):
        g = torch.Generator().manual_seed(2)
        H, N, C = 6, 40, 5
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1)
        D = torch.rand(H, C, C, generator=g) + 0.1
        part = ops.pi_hat_partial(D, preds, chunk_h=2)
        want = torch.einsum("hcs,hns->hnc", D, preds).sum(0)
        torch.testing.assert_close(part, want, rtol=1e-4, atol=1e-5)
        pi_xi, pi = ops.pi_hat_normalize(part)
        torch.testing.assert_close(pi_xi.sum(-1),
                                   torch.ones(N), rtol=1e-5, atol=1e-5)
        assert abs(float(pi.sum()) - 1.0) < 1e-5


class TestHypotheticalAndEig:
    def test_hypothetical_betas_dense(self):
        g = torch.Generator().manual_seed(3)
        H, C, B = 5, 4, 7
        a0 = torch.rand(H, C, generator=g) + 1
        b0 = torch.rand(H, C, generator=g) + 1
        cls = torch.randint(0, C, (B, H), generator=g)
        a, b = ops.hypothetical_betas(a0, b0, cls, 1.0)
        assert a.shape == (B, C, H)
        for bb in range(B):
            for c in range(C):
                for h in range(H):
                    da = 1.0 if cls[bb, h] == c else 0.0
                    assert abs(a[bb, c, h] - (a0[h, c] + da)) < 1e-6
                    assert abs(b[bb, c, h] - (b0[h, c] + 1 - da)) < 1e-6

    def test_eig_chunk_vs_composed(self):
        """ops.eig_chunk (the fused op) == composing the individual ops."""
        g = torch.Generator().manual_seed(4)
        H, C, B = 6, 4, 9
        a0 = torch.rand(H, C, generator=g) * 5 + 1
        b0 = torch.rand(H, C, generator=g) * 5 + 1
        cls = torch.randint(0, C, (B, H), generator=g)
        pi = torch.softmax(torch.rand(C, generator=g), 0)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1)

        pbest_before = ops.pbest_from_beta(a0.t().contiguous(),
                                           b0.t().contiguous())
        mixture0, H0 = ops.mixture_entropy(pbest_before, pi)
        got = ops.eig_chunk(a0, b0, cls, pbest_before, pi, pi_xi,
                            mixture0, H0)

        a, b = ops.hypothetical_betas(a0, b0, cls, 1.0)
        ph = ops.pbest_from_beta(a.reshape(B * C, H),
                                 b.reshape(B * C, H)).reshape(B, C, H)
        want = R.eig_assemble(ph, pbest_before, pi, pi_xi, mixture0, H0)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


class TestAcquisitionOps:
    def test_disagreement_mask_vs_mode(self):
        g = torch.Generator().manual_seed(5)
        cls = torch.randint(0, 3, (6, 100), generator=g)
        cls[:, :10] = 1  # force some all-agree points
        got = ops.disagreement_mask(cls)
        maj, _ = torch.mode(cls, dim=0)
        want = ((cls != maj).sum(0) > 0)
        torch.testing.assert_close(got, want)

    def test_accuracy_losses(self):
        g = torch.Generator().manual_seed(6)
        cls = torch.randint(0, 4, (5, 200), generator=g)
        labels = torch.randint(0, 4, (200,), generator=g)
        got = ops.accuracy_losses(cls, labels)
        want = 1 - (cls == labels).float().mean(1)
        torch.testing.assert_close(got, want)

    def test_entropy_acquisition(self):
        g = torch.Generator().manual_seed(7)
        p = torch.softmax(torch.randn(30, 5, generator=g), -1)
        got = ops.entropy_acquisition(p)
        want = -(p * torch.log(p + 1e-8)).sum(-1)
        torch.testing.assert_close(got, want)

    def test_vma_pairwise_vs_broadcast(self):
        g = torch.Generator().manual_seed(8)
        H, M = 9, 40
        losses = torch.rand(H, M, generator=g)
        got = ops.vma_pairwise(losses)
        diff = (losses.unsqueeze(0) - losses.unsqueeze(1)).abs()
        mask = torch.triu(torch.ones(H, H, dtype=torch.bool), diagonal=1)
        want = diff[mask].sum(0)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)

    def test_lure_weights_vs_loop(self):
        N, M = 500, 12
        g = torch.Generator().manual_seed(9)
        qs = torch.rand(M, generator=g) * 0.01 + 1e-4
        got = ops.lure_weights(qs, N)
        for m in range(M):
            m1 = m + 1
            v = 1 + ((N - M) / (N - m1)) * (1 / ((N - m1 + 1) * qs[m]) - 1)
            assert abs(got[m] - v) < 1e-3


class TestShardedEquivalence:
    """world==1 sharded ops must equal the plain ops exactly."""

    def test_pbest_sharded_single(self):
        from coda_amd.ops import sharded as S
        from coda_amd.parallel import Comm
        a, b = _rand_betas(rows=6, H=7, seed=11)
        got = S.pbest_from_beta_sharded(a, b, Comm())
        want = ops.pbest_from_beta(a, b)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-7)

    def test_eig_sharded_single(self):
        from coda_amd.ops import sharded as S
        from coda_amd.parallel import Comm
        g = torch.Generator().manual_seed(12)
        H, C, B = 5, 3, 6
        a0 = torch.rand(H, C, generator=g) * 5 + 1
        b0 = torch.rand(H, C, generator=g) * 5 + 1
        cls = torch.randint(0, C, (B, H), generator=g)
        pi = torch.softmax(torch.rand(C, generator=g), 0)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1)
        pb = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
        mixture0, H0 = ops.mixture_entropy(pb, pi)
        want = ops.eig_chunk(a0, b0, cls, pb, pi, pi_xi, mixture0, H0)
        comm = Comm()
        from coda_amd.ops.sharded import mixture_entropy_sharded
        m0s, H0s = mixture_entropy_sharded(pb, pi, comm)
        got = S.eig_chunk_sharded(a0, b0, cls, pb, pi, pi_xi, m0s, H0s, comm)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


class TestPiHatPacked:
    def test_packed_matches_fp32(self):
        g = torch.Generator().manual_seed(21)
        H, N, C = 6, 40, 5
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1)
        D = torch.rand(H, C, C, generator=g) + 0.1
        want = ops.pi_hat_partial(D, preds)
        packed = ops.pi_hat_pack(preds)
        assert packed.shape == (N, H * C) and packed.dtype == torch.bfloat16
        got = ops.pi_hat_partial_packed(D, packed)
        # bf16 inputs: ~0.4% relative tolerance
        torch.testing.assert_close(got, want, rtol=2e-2, atol=1e-3)

    def test_streamed_matches_packed(self):
        """N-chunked streamed pack+GEMM == the one-shot packed GEMM
        exactly (same bf16 rounding, including a ragged last chunk)."""
        g = torch.Generator().manual_seed(22)
        H, N, C = 5, 37, 4
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1)
        D = torch.rand(H, C, C, generator=g) + 0.1
        want = ops.pi_hat_partial_packed(D, ops.pi_hat_pack(preds))
        got = ops.pi_hat_partial_streamed(D, preds, chunk_n=16)
        torch.testing.assert_close(got, want)
        # bf16 storage input too
        got16 = ops.pi_hat_partial_streamed(D, preds.to(torch.bfloat16),
                                            chunk_n=16)
        torch.testing.assert_close(got16, want, rtol=2e-2, atol=1e-3)


class TestTableEig:
    """v2 (table-factored) EIG vs v1 (fused composition) equivalence."""

    def _setup(self, seed=31, H=7, C=5, B=11):
        g = torch.Generator().manual_seed(seed)
        a0 = torch.rand(H, C, generator=g) * 10 + 1
        b0 = torch.rand(H, C, generator=g) * 10 + 1
        cls = torch.randint(0, C, (B, H), generator=g)
        pi = torch.softmax(torch.rand(C, generator=g), 0)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1)
        pb = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
        m0, H0 = ops.mixture_entropy(pb, pi)
        return a0, b0, cls, pi, pi_xi, pb, m0, H0

    def test_pbest_hyp_matches_v1(self):
        from coda_amd.ops import table as T
        a0, b0, cls, pi, pi_xi, pb, m0, H0 = self._setup()
        B, H = cls.shape
        C = a0.shape[1]
        tables = T.table_precompute(a0, b0)
        got = T.pbest_hyp_table(tables, cls)
        ah, bh = ops.hypothetical_betas(a0, b0, cls, 1.0)
        want = ops.pbest_from_beta(ah.reshape(B * C, H),
                                   bh.reshape(B * C, H)).reshape(B, C, H)
        torch.testing.assert_close(got, want, rtol=2e-3, atol=1e-5)

    def test_eig_matches_v1(self):
        from coda_amd.ops import table as T
        a0, b0, cls, pi, pi_xi, pb, m0, H0 = self._setup(seed=32)
        tables = T.table_precompute(a0, b0)
        got = T.eig_chunk_table(tables, cls, pb, pi, pi_xi, m0, H0)
        want = ops.eig_chunk(a0, b0, cls, pb, pi, pi_xi, m0, H0)
        torch.testing.assert_close(got, want, rtol=5e-3, atol=1e-5)

    def test_concentrated_betas_finite(self):
        from coda_amd.ops import table as T
        g = torch.Generator().manual_seed(33)
        H, C, B = 5, 4, 6
        a0 = torch.rand(H, C, generator=g) * 3000 + 100
        b0 = torch.rand(H, C, generator=g) * 3000 + 100
        cls = torch.randint(0, C, (B, H), generator=g)
        tables = T.table_precompute(a0, b0)
        pb = T.pbest_hyp_table(tables, cls)
        assert torch.isfinite(pb).all()
        torch.testing.assert_close(pb.sum(-1), torch.ones(B, C), atol=2e-3,
                                   rtol=0)

    def test_selector_table_impl_matches_fused(self):
        import random
        from coda_amd import CODA
        from coda_amd.datasets import Dataset, make_synthetic_task
        preds, labels = make_synthetic_task(H=6, N=200, C=4, seed=34)
        ds = Dataset.from_tensors(preds, labels, "cpu")
        random.seed(0); torch.manual_seed(0)
        s1 = CODA(ds, chunk_size=64, eig_impl="fused")
        e1, c1 = s1.eig_batched()
        random.seed(0); torch.manual_seed(0)
        s2 = CODA(ds, chunk_size=64, eig_impl="table")
        e2, c2 = s2.eig_batched()
        assert c1 == c2
        torch.testing.assert_close(e1, e2, rtol=5e-3, atol=1e-5)

    def test_incremental_table_update_matches_rebuild(self):
        from coda_amd.ops import table as T
        g = torch.Generator().manual_seed(35)
        H, C = 6, 5
        a0 = torch.rand(H, C, generator=g) * 10 + 1
        b0 = torch.rand(H, C, generator=g) * 10 + 1
        tables = T.table_precompute(a0, b0)
        # perturb class rows 1 and 3 (what add_label does)
        a0[:, 1] += 0.01; b0[:, 3] += 0.02
        T.table_update_rows(tables, a0, b0, [1, 3])
        fresh = T.table_precompute(a0, b0)
        torch.testing.assert_close(tables.EG, fresh.EG)
        torch.testing.assert_close(tables.delta, fresh.delta)
        torch.testing.assert_close(tables.s_base, fresh.s_base)


def test_pbest_hchunked_matches_plain():
    """Wide-H chunked passes == plain eager (used beyond the kernel's
    LDS budget, H > 2048)."""
    a, b = _rand_betas(rows=5, H=37, seed=51)
    got = R.pbest_from_beta_hchunked(a, b, chunk_h=8)
    want = R.pbest_from_beta(a, b)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-6)


def test_wide_h_dispatch_path():
    """ops.pbest_from_beta routes H > 2048 through the chunked passes."""
    a, b = _rand_betas(rows=2, H=2100, seed=52, lo=1.0, hi=20.0)
    p = ops.pbest_from_beta(a, b)
    assert torch.isfinite(p).all()
    np.testing.assert_allclose(p.sum(-1).numpy(), 1.0, atol=1e-3)
