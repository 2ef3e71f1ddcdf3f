"""Property-based tests of the compute core (hypothesis)."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from coda_amd import ops


def _betas(draw, rows, H, lo=0.6, hi=80.0):
    a = draw(st.lists(st.floats(lo, hi), min_size=rows * H,
                      max_size=rows * H))
    b = draw(st.lists(st.floats(lo, hi), min_size=rows * H,
                      max_size=rows * H))
    return (torch.tensor(a, dtype=torch.float32).view(rows, H),
            torch.tensor(b, dtype=torch.float32).view(rows, H))


@settings(max_examples=20, deadline=None)
@given(st.data())
def test_pbest_permutation_equivariance(data):
    """Permuting the model axis permutes P(best) identically."""
    a, b = _betas(data.draw, rows=3, H=5)
    p = ops.pbest_from_beta(a, b)
    perm = torch.randperm(5)
    p2 = ops.pbest_from_beta(a[:, perm].contiguous(),
                             b[:, perm].contiguous())
    torch.testing.assert_close(p2, p[:, perm], rtol=1e-4, atol=1e-6)


@settings(max_examples=20, deadline=None)
@given(st.data())
def test_pbest_valid_distribution(data):
    a, b = _betas(data.draw, rows=4, H=6)
    p = ops.pbest_from_beta(a, b)
    assert torch.isfinite(p).all()
    assert (p >= 0).all()
    torch.testing.assert_close(p.sum(-1), torch.ones(4), atol=2e-3, rtol=0)


@settings(max_examples=15, deadline=None)
@given(st.floats(1.0, 50.0), st.floats(1.0, 50.0), st.floats(0.5, 20.0))
def test_pbest_monotone_in_alpha(a0, b0, bump):
    """Raising one model's alpha (more successes) raises its P(best)."""
    a = torch.tensor([[a0, a0]], dtype=torch.float32)
    b = torch.tensor([[b0, b0]], dtype=torch.float32)
    p_eq = ops.pbest_from_beta(a, b)[0]
    a2 = a.clone()
    a2[0, 0] += bump
    p_up = ops.pbest_from_beta(a2, b)[0]
    assert p_up[0] > p_eq[0] - 1e-6


@settings(max_examples=15, deadline=None)
@given(st.integers(2, 9), st.integers(5, 40), st.integers(0, 9999))
def test_vma_identity_random(H, M, seed):
    g = torch.Generator().manual_seed(seed)
    losses = torch.rand(H, M, generator=g)
    got = ops.vma_pairwise(losses)
    diff = (losses.unsqueeze(0) - losses.unsqueeze(1)).abs()
    mask = torch.triu(torch.ones(H, H, dtype=torch.bool), diagonal=1)
    want = diff[mask].sum(0)
    torch.testing.assert_close(got, want, rtol=1e-3, atol=1e-4)


@settings(max_examples=10, deadline=None)
@given(st.integers(2, 6), st.integers(3, 8), st.integers(2, 7),
       st.integers(0, 9999))
def test_table_matches_fused_random(H, C, B, seed):
    """v2 tables == composed v1 on random shapes."""
    from coda_amd.ops import table as T
    g = torch.Generator().manual_seed(seed)
    a0 = torch.rand(H, C, generator=g) * 30 + 1
    b0 = torch.rand(H, C, generator=g) * 30 + 1
    cls = torch.randint(0, C, (B, H), generator=g)
    tables = T.table_precompute(a0, b0)
    got = T.pbest_hyp_table(tables, cls)
    ah, bh = ops.hypothetical_betas(a0, b0, cls, 1.0)
    want = ops.pbest_from_beta(ah.reshape(B * C, H),
                               bh.reshape(B * C, H)).reshape(B, C, H)
    torch.testing.assert_close(got, want, rtol=5e-3, atol=1e-4)


@settings(max_examples=15, deadline=None)
@given(st.integers(2, 8), st.integers(5, 60), st.integers(2, 9),
       st.integers(1, 64), st.integers(0, 9999))
def test_streamed_pi_hat_random(H, N, C, chunk_n, seed):
    """N-chunked streamed pack+GEMM == one-shot packed GEMM for any
    chunking (including chunk_n > N and ragged tails)."""
    g = torch.Generator().manual_seed(seed)
    preds = torch.softmax(torch.randn(H, N, C, generator=g), -1)
    D = torch.rand(H, C, C, generator=g) + 0.05
    want = ops.pi_hat_partial_packed(D, ops.pi_hat_pack(preds))
    got = ops.pi_hat_partial_streamed(D, preds, chunk_n=chunk_n)
    torch.testing.assert_close(got, want)
