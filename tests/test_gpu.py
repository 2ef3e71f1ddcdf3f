"""GPU tests: HIP kernels vs eager goldens, end-to-end CODA on ROCm.

All marked `gpu`; run on an MI355X box with the in-tree extension built
(`python build_hip.py`). The kernels evaluate the Beta log-pdf in f64, so
they are checked BOTH against the fp32 eager ops (loose) and against the
fp64 NumPy golden (tight - the kernel should be closer to fp64 truth than
eager fp32 is).
"""
import random

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    import coda_amd.ops as ops
    assert ops.hip_available(), \
        "HIP extension must be built on GPU boxes (python build_hip.py)"


def _rand_betas(rows, H, seed=0, lo=0.5, hi=60.0, device="cpu"):
    g = torch.Generator().manual_seed(seed)
    a = torch.rand(rows, H, generator=g) * (hi - lo) + lo
    b = torch.rand(rows, H, generator=g) * (hi - lo) + lo
    return a.to(device), b.to(device)


class TestPbestKernel:
    def test_vs_eager(self, dev):
        from coda_amd import ops
        a, b = _rand_betas(16, 24, seed=0, device=dev)
        got = ops.pbest_from_beta(a, b).cpu()
        want = ops.reference.pbest_from_beta(a.cpu(), b.cpu())
        torch.testing.assert_close(got, want, rtol=2e-3, atol=1e-5)

    def test_vs_fp64_golden(self, dev):
        from coda_amd import ops
        from tests.test_ops import _pbest_fp64
        a, b = _rand_betas(8, 12, seed=1, device=dev)
        got = ops.pbest_from_beta(a, b).cpu().numpy()
        want = _pbest_fp64(a.cpu(), b.cpu())
        np.testing.assert_allclose(got, want, rtol=5e-4, atol=5e-6)

    def test_large_params_finite_and_accurate(self, dev):
        """Concentrated Betas: the kernel's f64 log-pdf path should beat
        eager fp32 against the fp64 golden."""
        from coda_amd import ops
        from tests.test_ops import _pbest_fp64
        a, b = _rand_betas(4, 8, seed=2, lo=500.0, hi=20000.0, device=dev)
        got = ops.pbest_from_beta(a, b).cpu().numpy()
        assert np.isfinite(got).all()
        want = _pbest_fp64(a.cpu(), b.cpu())
        np.testing.assert_allclose(got, want, rtol=5e-3, atol=1e-5)

    def test_rows_sum_to_one(self, dev):
        from coda_amd import ops
        a, b = _rand_betas(64, 128, seed=3, device=dev)
        p = ops.pbest_from_beta(a, b)
        np.testing.assert_allclose(p.sum(-1).cpu().numpy(), 1.0, atol=1e-3)

    def test_wide_H(self, dev):
        from coda_amd import ops
        a, b = _rand_betas(4, 1024, seed=4, device=dev)
        p = ops.pbest_from_beta(a, b)
        assert torch.isfinite(p).all()
        np.testing.assert_allclose(p.sum(-1).cpu().numpy(), 1.0, atol=1e-3)


class TestEigKernel:
    def test_vs_eager(self, dev):
        from coda_amd import ops
        g = torch.Generator().manual_seed(5)
        H, C, B = 16, 10, 32
        a0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        b0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        cls = torch.randint(0, C, (B, H), generator=g).to(dev)
        pi = torch.softmax(torch.rand(C, generator=g), 0).to(dev)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1).to(dev)

        pb = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
        m0, H0 = ops.mixture_entropy(pb, pi)
        got = ops.eig_chunk(a0, b0, cls, pb, pi, pi_xi, m0, H0).cpu()

        want = ops.reference.pbest_from_beta(
            *(t.cpu() for t in (a0.t().contiguous(), b0.t().contiguous())))
        m0c, H0c = ops.reference.mixture_entropy(want, pi.cpu())
        ah, bh = ops.reference.hypothetical_betas(a0.cpu(), b0.cpu(),
                                                  cls.cpu(), 1.0)
        ph = ops.reference.pbest_from_beta(
            ah.reshape(B * C, H), bh.reshape(B * C, H)).reshape(B, C, H)
        want_eig = ops.reference.eig_assemble(ph, want, pi.cpu(),
                                              pi_xi.cpu(), m0c, H0c)
        torch.testing.assert_close(got, want_eig, rtol=5e-3, atol=1e-4)


class TestEndToEnd:
    def test_coda_gpu_trajectory_matches_cpu(self, dev):
        """Same seeds, GPU (HIP kernels) vs CPU (eager): identical
        selections over 5 steps (tolerances inside the selector are loose
        enough that fp32-vs-kernel differences don't flip argmaxes on this
        well-separated task)."""
        from coda_amd import CODA, Oracle
        from coda_amd.datasets import Dataset, make_synthetic_task
        from coda_amd.options import LOSS_FNS

        preds, labels = make_synthetic_task(H=8, N=300, C=5, seed=0)

        def run(device):
            ds = Dataset.from_tensors(preds, labels, device)
            oracle = Oracle(ds, LOSS_FNS["acc"])
            random.seed(0); torch.manual_seed(0)
            sel = CODA(ds, chunk_size=64, pi_hat_precision="fp32")
            traj = []
            for _ in range(5):
                idx, q = sel.get_next_item_to_label()
                sel.add_label(idx, oracle(int(idx)), q)
                traj.append((int(idx),
                             int(sel.get_best_model_prediction())))
            return traj, sel.get_pbest().cpu()

        t_cpu, p_cpu = run("cpu")
        t_gpu, p_gpu = run(dev)
        assert t_cpu == t_gpu
        torch.testing.assert_close(p_cpu, p_gpu, rtol=2e-3, atol=1e-4)

    def test_smoke_entry(self):
        import __graft_entry__
        __graft_entry__.smoke()

    def test_ops_fail_loudly_without_ext(self, dev, monkeypatch):
        """On a GPU box with the extension 'missing', hot ops must raise."""
        import coda_amd.ops as ops
        monkeypatch.setattr(ops, "_ext", None)
        monkeypatch.setattr(ops, "_ext_err", ImportError("simulated"))
        monkeypatch.delenv("CODA_AMD_ALLOW_EAGER", raising=False)
        a, b = _rand_betas(2, 4, device=dev)
        with pytest.raises(RuntimeError, match="HIP extension"):
            ops.pbest_from_beta(a, b)


class TestShardedPhases:
    """Single-GPU equivalence: the two-phase sharded kernels composed with
    a no-op Comm must match the fused kernels (this chains with the CPU
    gloo test: eager-sharded == eager, phases == fused, so
    HIP-sharded-on-N-ranks inherits correctness)."""

    def test_pbest_phases_equal_fused(self, dev):
        from coda_amd import ops
        from coda_amd.ops import sharded as S
        from coda_amd.parallel import Comm
        a, b = _rand_betas(32, 48, seed=21, device=dev)
        got = S.pbest_from_beta_sharded(a, b, Comm())
        want = ops.pbest_from_beta(a, b)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-6)

    def test_eig_phases_equal_fused(self, dev):
        from coda_amd import ops
        from coda_amd.ops import sharded as S
        from coda_amd.parallel import Comm
        g = torch.Generator().manual_seed(22)
        H, C, B = 16, 10, 24
        a0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        b0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        cls = torch.randint(0, C, (B, H), generator=g).to(dev)
        pi = torch.softmax(torch.rand(C, generator=g), 0).to(dev)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1).to(dev)
        pb = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
        m0, H0 = ops.mixture_entropy(pb, pi)
        want = ops.eig_chunk(a0, b0, cls, pb, pi, pi_xi, m0, H0)
        comm = Comm()
        m0s, H0s = S.mixture_entropy_sharded(pb, pi, comm)
        got = S.eig_chunk_sharded(a0, b0, cls, pb, pi, pi_xi, m0s, H0s,
                                  comm)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


class TestStorageGPU:
    def test_streamed_ingest_matches_direct(self, dev, tmp_path):
        from coda_amd.datasets import Dataset, write_synthetic_task
        path = write_synthetic_task(str(tmp_path), name="st", H=6, N=400,
                                    C=8, seed=11)
        direct = Dataset(path, dev, storage_dtype="bf16")
        streamed = Dataset(path, dev, storage_dtype="bf16",
                           stream_chunk_mb=1)
        torch.cuda.synchronize()
        assert torch.equal(direct.preds, streamed.preds)

    def test_fp8_storage_coda_runs(self, dev, tmp_path):
        from coda_amd import CODA, Oracle
        from coda_amd.datasets import Dataset, write_synthetic_task
        from coda_amd.options import LOSS_FNS
        path = write_synthetic_task(str(tmp_path), name="f8", H=6, N=300,
                                    C=5, seed=12, best_acc=0.95,
                                    worst_acc=0.4)
        ds = Dataset(path, dev, storage_dtype="fp8")
        assert ds.preds.dtype == torch.float8_e4m3fn
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=64)
        for _ in range(3):
            i, q = sel.get_next_item_to_label()
            sel.add_label(i, oracle(int(i)), q)
        p = sel.get_pbest()
        assert torch.isfinite(p).all()
        assert abs(float(p.sum()) - 1.0) < 1e-3


class TestTableFusionKernels:
    def test_eig_chunk_table_gpu_matches_cpu(self, dev, monkeypatch):
        """The es_build/eig_assemble_k fused GPU path == the torch
        composition on CPU (fp32 GEMM for exact comparison; the default
        bf16 GEMM is checked separately with an absolute tolerance)."""
        monkeypatch.setenv("CODA_AMD_V2_GEMM", "fp32")
        from coda_amd import ops
        from coda_amd.ops import table as T
        g = torch.Generator().manual_seed(41)
        H, C, B = 16, 10, 24
        a0 = torch.rand(H, C, generator=g) * 20 + 1
        b0 = torch.rand(H, C, generator=g) * 20 + 1
        cls = torch.randint(0, C, (B, H), generator=g)
        pi = torch.softmax(torch.rand(C, generator=g), 0)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1)
        pb0 = ops.reference.pbest_from_beta(a0.t().contiguous(),
                                            b0.t().contiguous())
        m0, H0 = ops.reference.mixture_entropy(pb0, pi)

        t_cpu = T.table_precompute(a0, b0)
        want = T.eig_chunk_table(t_cpu, cls, pb0, pi, pi_xi, m0, H0)

        t_gpu = T.table_precompute(a0.to(dev), b0.to(dev))
        assert t_gpu.eg16 is None  # fp32 mode
        got = T.eig_chunk_table(t_gpu, cls.to(dev), pb0.to(dev),
                                pi.to(dev), pi_xi.to(dev), m0.to(dev),
                                H0.to(dev)).cpu()
        torch.testing.assert_close(got, want, rtol=5e-3, atol=1e-5)

        # default bf16 pairing GEMM: absolute accuracy at the fp32
        # reduction-noise level (EIG entropies are insensitive to the
        # 0.4% per-element input rounding; measured ~5e-7)
        monkeypatch.delenv("CODA_AMD_V2_GEMM")
        t16 = T.table_precompute(a0.to(dev), b0.to(dev))
        assert t16.eg16 is not None
        got16 = T.eig_chunk_table(t16, cls.to(dev), pb0.to(dev),
                                  pi.to(dev), pi_xi.to(dev), m0.to(dev),
                                  H0.to(dev)).cpu()
        assert float((got16 - want).abs().max()) < 5e-5

    def test_sharded_table_gpu_path_single_rank(self, dev):
        """es_build_gathered + eig_totals/eig_entropy with a no-op Comm
        must match the single-device fused path."""
        from coda_amd import ops
        from coda_amd.ops import table as T
        from coda_amd.parallel import Comm
        g = torch.Generator().manual_seed(42)
        H, C, B = 16, 10, 24
        a0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        b0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        cls = torch.randint(0, C, (B, H), generator=g).to(dev)
        pi = torch.softmax(torch.rand(C, generator=g), 0).to(dev)
        pi_xi = torch.softmax(torch.rand(B, C, generator=g), -1).to(dev)
        pb0 = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
        m0, H0 = ops.mixture_entropy(pb0, pi)
        tables = T.table_precompute(a0, b0)
        want = T.eig_chunk_table(tables, cls, pb0, pi, pi_xi, m0, H0)
        comm = Comm()
        sba = T.s_base_global(tables, comm)
        got = T.eig_chunk_table_sharded(tables, sba, cls, pb0, pi, pi_xi,
                                        m0, H0, comm)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-6)

    def test_beta_row_tables_kernel_matches_precompute(self, dev):
        from coda_amd.ops import table as T
        g = torch.Generator().manual_seed(43)
        H, C = 12, 6
        a0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        b0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
        tables = T.table_precompute(a0, b0)
        a0[:, 2] += 0.01
        T.table_update_rows(tables, a0, b0, [2])  # kernel path on GPU
        fresh = T.table_precompute(a0, b0)
        # kernel cdf comes from the wave scan, precompute's from cumsum:
        # fp32 association differences up to ~1e-3 absolute are expected
        torch.testing.assert_close(tables.EG, fresh.EG, rtol=2e-3,
                                   atol=1e-2)
        torch.testing.assert_close(tables.delta, fresh.delta, rtol=1e-3,
                                   atol=5e-3)
        torch.testing.assert_close(tables.s_base, fresh.s_base,
                                   rtol=1e-4, atol=1e-2)

    def test_pi_hat_delta_kernel(self, dev):
        from coda_amd import ops
        g = torch.Generator().manual_seed(44)
        H, N, C = 9, 500, 7
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1).to(dev)
        cls = torch.randint(0, C, (H,), generator=g).to(dev)
        got = ops.pi_hat_delta(preds, cls)
        want = ops.reference.pi_hat_delta(preds.cpu(), cls.cpu()).to(dev)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
        # bf16 storage
        got16 = ops.pi_hat_delta(preds.to(torch.bfloat16).contiguous(), cls)
        torch.testing.assert_close(got16, want, rtol=2e-2, atol=1e-2)
        # fp8 storage (e4m3): compare against the eager op on the SAME
        # fp8 values (the storage rounding is semantics, not kernel error)
        p8 = preds.to(torch.float8_e4m3fn).contiguous()
        got8 = ops.pi_hat_delta(p8, cls)
        want8 = ops.reference.pi_hat_delta(p8.cpu(), cls.cpu()).to(dev)
        torch.testing.assert_close(got8, want8, rtol=1e-3, atol=1e-3)

    def test_dirichlet_add_kernel(self, dev):
        from coda_amd import ops
        g = torch.Generator().manual_seed(48)
        H, C = 37, 11
        dir0 = torch.rand(H, C, C, generator=g).to(dev)
        cls = torch.randint(0, C, (H,), generator=g).to(dev)
        y = torch.tensor([4], dtype=torch.long, device=dev)
        lr = 0.01
        want = dir0.clone()
        onehot = torch.nn.functional.one_hot(cls, C).float()
        want.index_add_(1, y, (lr * onehot).unsqueeze(1))
        ops._ext.dirichlet_add(dir0, y, cls, lr)
        torch.testing.assert_close(dir0, want)

    def test_col_add_kernel(self, dev):
        from coda_amd import ops
        g = torch.Generator().manual_seed(47)
        N, C = 1234, 56
        A = torch.rand(N, C, generator=g).to(dev)
        rs = torch.rand(N, generator=g).to(dev)
        delta = torch.randn(N, generator=g).to(dev)
        y = torch.tensor([13], dtype=torch.long, device=dev)
        A2, rs2 = A.clone(), rs.clone()
        A2.index_add_(1, y, delta.unsqueeze(1))
        rs2 += delta
        ops._ext.col_add(A, rs, y, delta)
        torch.testing.assert_close(A, A2)
        torch.testing.assert_close(rs, rs2)

    def test_pi_marginal_kernel(self, dev):
        from coda_amd import ops
        g = torch.Generator().manual_seed(45)
        N, C = 777, 133
        A = (torch.rand(N, C, generator=g) + 0.01).to(dev)
        rs = A.sum(-1)
        got = ops._ext.pi_marginal(A, rs)
        want = (A / rs.clamp_min(1e-12).unsqueeze(-1)).sum(0)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


class TestWidePoolGPU:
    def test_hchunked_pbest_gpu(self, dev):
        """H beyond the fused kernel's LDS budget routes through the
        two-pass window kernels; matches the chunked eager math."""
        from coda_amd import ops
        a, b = _rand_betas(rows=3, H=2100, seed=61, lo=1.0, hi=20.0,
                           device=dev)
        p = ops.pbest_from_beta(a, b)
        assert torch.isfinite(p).all()
        np.testing.assert_allclose(p.sum(-1).cpu().numpy(), 1.0, atol=1e-3)
        want = ops.reference.pbest_from_beta_hchunked(a.cpu(), b.cpu())
        torch.testing.assert_close(p.cpu(), want, rtol=2e-3, atol=1e-5)

    def test_wide_pbest_equals_fused_at_boundary(self, dev):
        """Window kernels (forced) == fused kernel on an H that both
        support: same math, different coupling route."""
        from coda_amd import ops
        a, b = _rand_betas(rows=5, H=1500, seed=62, lo=0.5, hi=40.0,
                           device=dev)
        fused = ops.pbest_from_beta(a, b)
        wide = ops._pbest_wide_hip(a, b, hc=512)
        torch.testing.assert_close(wide, fused, rtol=1e-4, atol=1e-6)
        # uneven tail window (1500 = 2*640 + 220)
        wide2 = ops._pbest_wide_hip(a, b, hc=640)
        torch.testing.assert_close(wide2, fused, rtol=1e-4, atol=1e-6)

    def test_pi_hat_delta_wide_pool(self, dev):
        """H-chunked pi_hat_delta partials (H > 512 dispatch) match the
        eager reference."""
        from coda_amd import ops
        g = torch.Generator().manual_seed(46)
        H, N, C = 600, 300, 7
        preds = torch.softmax(torch.randn(H, N, C, generator=g), -1).to(dev)
        cls = torch.randint(0, C, (H,), generator=g).to(dev)
        got = ops.pi_hat_delta(preds, cls)
        want = ops.reference.pi_hat_delta(preds.cpu(), cls.cpu()).to(dev)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
        got16 = ops.pi_hat_delta(preds.to(torch.bfloat16).contiguous(), cls)
        torch.testing.assert_close(got16, want, rtol=2e-2, atol=1e-2)

    def test_coda_wide_pool_gpu(self, dev):
        """End-to-end CODA on a 2100-model pool (wide-H fallback for the
        class-row posterior + v2 tables for EIG)."""
        import bench
        from coda_amd import CODA, Oracle
        from coda_amd.datasets import Dataset
        from coda_amd.options import LOSS_FNS
        preds, labels = bench.synth_preds(list(range(2100)), 500, 10, dev)
        ds = Dataset.from_tensors(preds, labels, dev)
        oracle = Oracle(ds, LOSS_FNS["acc"])
        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds, chunk_size=64, prefilter_n=64)
        for _ in range(2):
            i, q = sel.get_next_item_to_label()
            sel.add_label(i, oracle(int(i)), q)
        p = sel.get_pbest()
        assert p.shape == (2100,) and torch.isfinite(p).all()


class TestLabelGraph:
    def test_graphed_trajectory_equals_eager(self, dev, monkeypatch):
        """hipGraph label-update replay == the eager pipeline: identical
        selections and P(best) over 6 steps."""
        import bench
        from coda_amd import CODA, Oracle
        from coda_amd.datasets import Dataset
        from coda_amd.options import LOSS_FNS

        preds, labels = bench.synth_preds(list(range(12)), 600, 8, dev)
        ds = Dataset.from_tensors(preds, labels, dev)
        oracle = Oracle(ds, LOSS_FNS["acc"])

        # production mode for both the selector gate and the op guards
        import coda_amd.selectors.coda as coda_mod
        import coda_amd.ops as ops_mod
        import coda_amd.util as util_mod
        monkeypatch.setattr(coda_mod, "DEBUG", False)
        monkeypatch.setattr(ops_mod, "DEBUG", False)
        monkeypatch.setattr(util_mod, "DEBUG", False)

        def run(no_graph):
            if no_graph:
                monkeypatch.setenv("CODA_AMD_NO_GRAPH", "1")
            else:
                monkeypatch.delenv("CODA_AMD_NO_GRAPH", raising=False)
            random.seed(0); torch.manual_seed(0)
            sel = CODA(ds, chunk_size=64, prefilter_n=64)
            traj = []
            for _ in range(6):
                i, q = sel.get_next_item_to_label()
                sel.add_label(i, oracle(int(i)), q)
                traj.append((int(i), int(sel.get_best_model_prediction())))
            assert (sel._label_graph is None) == no_graph
            return traj, sel.get_pbest().cpu(), sel.pi_hat.cpu()

        t_e, p_e, pi_e = run(True)
        t_g, p_g, pi_g = run(False)
        assert t_e == t_g
        torch.testing.assert_close(p_e, p_g, rtol=1e-5, atol=1e-7)
        torch.testing.assert_close(pi_e, pi_g, rtol=1e-5, atol=1e-7)


class TestPairKernels:
    """v3 pair engine kernels (ops/hip/pair.hip) vs the eager pair math."""

    def test_mfma_fragment_layout(self, dev):
        """16x16x32 bf16 MFMA probe: A (16,32) x B^T rows (16,32) vs a
        bf16-rounded torch matmul. Asymmetric operands catch row/col
        swaps (guide section 3)."""
        from coda_amd import ops
        g = torch.Generator().manual_seed(7)
        a = torch.randn(16, 32, generator=g).to(dev)
        bt = (torch.arange(16 * 32, dtype=torch.float32).reshape(16, 32)
              / 100.0 - 2.0).to(dev)
        got = ops._ext.mfma_probe(a, bt).cpu()
        want = (a.cpu().to(torch.bfloat16).float()
                @ bt.cpu().to(torch.bfloat16).float().t())
        torch.testing.assert_close(got, want, rtol=1e-2, atol=1e-2)

    def test_pair_kernels_vs_eager(self, dev):
        """pair_dsum_es + pair_gemm_entropy vs the fp32 eager pair
        formulation, through the full eig_pairs dispatch."""
        from coda_amd.ops import pair as pops
        from coda_amd.ops import table as tops
        from coda_amd.ops import reference as R
        from tests.test_pair import _random_problem
        for H, N, C in [(10, 120, 7), (3, 60, 2), (16, 200, 126),
                        (128, 500, 50)]:
            (preds, cls, dirichlets, pi_hat, adjusted,
             row_sums) = _random_problem(H, N, C, seed=H + C)
            alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
            alpha_cc, beta_cc = alpha_cc.to(dev), beta_cc.to(dev)
            tables = tops.table_precompute(alpha_cc, beta_cc)
            tables = pops.attach_pair_tables(tables)
            pbest_before = R.pbest_from_beta(
                alpha_cc.t().contiguous(), beta_cc.t().contiguous())
            mixture0, H_before = R.mixture_entropy(
                pbest_before, pi_hat.to(dev))
            ids = torch.arange(N, device=dev)
            cls_rows = cls.to(dev)[:, ids].t().to(torch.int32).contiguous()
            ps = pops.build_pairs(cls_rows, ids, C)
            eig_k = pops.eig_pairs(
                tables, ps, cls_rows, pbest_before, pi_hat.to(dev),
                mixture0, H_before, adjusted.to(dev), row_sums.to(dev))
            h_eager = pops.pair_h_after(
                tables, ps, cls_rows, pbest_before, pi_hat.to(dev),
                mixture0)
            eig_e = pops.eig_from_pairs(
                h_eager, ps, adjusted.to(dev), row_sums.to(dev),
                H_before)[ps.cand_ids]
            # bf16 GEMM operands; EIG differences are where the signal
            # (h_after - h_base) loses a digit to bf16 rounding, so the
            # tolerance is at the EIG-value scale with an absolute floor
            # for near-zero-EIG shapes (round-1 calibration: max rel
            # ~6e-4 at O(1) scales, abs ~2e-4 worst case)
            scale = float(eig_e.abs().max())
            err = float((eig_k - eig_e).abs().max())
            assert err < max(5e-3 * scale, 3e-4), (H, N, C, err, scale)

    def test_pair_trajectory_matches_table(self, dev):
        """Full selector on cuda: eig_impl='pair' (the GPU default) vs
        'table' over the no-prefilter acquisition. Both run bf16
        pairing GEMMs with different factorizations, so candidates
        whose EIG values are within the bf16 noise band may swap rank -
        selections must agree except at such near-ties, and the EIG
        vectors themselves must agree to GEMM tolerance."""
        import bench
        from coda_amd import CODA, Oracle
        from coda_amd.datasets import Dataset
        from coda_amd.options import LOSS_FNS

        preds, labels = bench.synth_preds(list(range(16)), 800, 12, dev)
        ds = Dataset.from_tensors(preds, labels, dev)
        oracle = Oracle(ds, LOSS_FNS["acc"])

        def run(impl):
            random.seed(0); torch.manual_seed(0)
            sel = CODA(ds, eig_impl=impl)
            traj, qs, cands = [], [], []
            for _ in range(8):
                q_vals, cand = sel.eig_batched()
                qs.append(q_vals.detach().clone())
                cands.append(list(cand))
                i, q = sel.get_next_item_to_label()
                sel.add_label(i, oracle(int(i)), q)
                traj.append(int(i))
            return traj, qs, cands

        t_t, q_t, c_t = run("table")
        t_p, q_p, c_p = run("pair")
        for s in range(8):
            if t_t[s] == t_p[s]:
                assert c_t[s] == c_p[s]
                scale = float(q_t[s].max())
                assert float((q_t[s] - q_p[s]).abs().max()) < 3e-3 * scale
                continue
            # first divergence: must be a near-tie in the table engine's
            # own values; states differ beyond this step - stop
            pos_p = c_t[s].index(t_p[s])
            pos_t = c_t[s].index(t_t[s])
            gap = float((q_t[s][pos_t] - q_t[s][pos_p]).abs())
            assert gap < 2e-3 * float(q_t[s].max()), \
                (s, t_t[s], t_p[s], gap)
            break

    def test_pair_wide_pipeline_vs_eager(self, dev):
        """288 < 2H: the split wide pipeline (M-writing 128x128 GEMM +
        wave-per-pair entropy) vs the eager pair math."""
        from coda_amd.ops import pair as pops
        from coda_amd.ops import table as tops
        from coda_amd.ops import reference as R
        from tests.test_pair import _random_problem
        H, N, C = 160, 300, 30
        (preds, cls, dirichlets, pi_hat, adjusted,
         row_sums) = _random_problem(H, N, C, seed=7)
        alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
        alpha_cc, beta_cc = alpha_cc.to(dev), beta_cc.to(dev)
        tables = pops.attach_pair_tables(
            tops.table_precompute(alpha_cc, beta_cc))
        pbest_before = R.pbest_from_beta(alpha_cc.t().contiguous(),
                                         beta_cc.t().contiguous())
        mixture0, H_before = R.mixture_entropy(pbest_before,
                                               pi_hat.to(dev))
        ids = torch.arange(N, device=dev)
        cls_rows = cls.to(dev)[:, ids].t().to(torch.int32).contiguous()
        ps = pops.build_pairs(cls_rows, ids, C, tile=128)
        assert 2 * H > 288  # wide path engaged
        eig_k = pops.eig_pairs(
            tables, ps, cls_rows, pbest_before, pi_hat.to(dev),
            mixture0, H_before, adjusted.to(dev), row_sums.to(dev))
        h_eager = pops.pair_h_after(tables, ps, cls_rows, pbest_before,
                                    pi_hat.to(dev), mixture0)
        eig_e = pops.eig_from_pairs(h_eager, ps, adjusted.to(dev),
                                    row_sums.to(dev),
                                    H_before)[ps.cand_ids]
        scale = float(eig_e.abs().max())
        err = float((eig_k - eig_e).abs().max())
        assert err < max(5e-3 * scale, 3e-4), (err, scale)

    def test_pair_wide_cls_route_vs_table(self, dev):
        """H > 1024 (the 10k-pool regime): the vmask-less cls-based pair
        route vs the v2 table engine on the same chunk."""
        from coda_amd.ops import pair as pops
        from coda_amd.ops import table as tops
        from coda_amd.ops import reference as R
        from tests.test_pair import _random_problem
        H, N, C = 1200, 200, 25
        (preds, cls, dirichlets, pi_hat, adjusted,
         row_sums) = _random_problem(H, N, C, seed=11)
        alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
        alpha_cc, beta_cc = alpha_cc.to(dev), beta_cc.to(dev)
        tables = pops.attach_pair_tables(
            tops.table_precompute(alpha_cc, beta_cc))
        from coda_amd import ops as O
        pbest_before = O.pbest_from_beta(alpha_cc.t().contiguous(),
                                         beta_cc.t().contiguous())
        mixture0, H_before = R.mixture_entropy(pbest_before,
                                               pi_hat.to(dev))
        ids = torch.arange(N, device=dev)
        cls_rows = cls.to(dev)[:, ids].t().to(torch.int32).contiguous()
        ps = pops.build_pairs(cls_rows, ids, C, tile=128,
                              with_vmask=False)
        assert ps.vmask is None
        eig_k = pops.eig_pairs(
            tables, ps, cls_rows, pbest_before, pi_hat.to(dev),
            mixture0, H_before, adjusted.to(dev), row_sums.to(dev))
        pi_xi = (adjusted / row_sums.clamp_min(1e-12).unsqueeze(-1)) \
            .to(dev)
        eig_t = tops.eig_chunk_table(tables, cls_rows.long(),
                                     pbest_before, pi_hat.to(dev),
                                     pi_xi, mixture0, H_before)
        scale = float(eig_t.abs().max())
        err = float((eig_k - eig_t).abs().max())
        assert err < max(5e-3 * scale, 3e-4), (err, scale)

    def test_checkpoint_restore_with_pair_engine(self, dev):
        """Mid-run checkpoint/restore on GPU: the pair structures,
        hipGraphs and replicated caches must rebuild cleanly and the
        resumed trajectory must continue identically."""
        import bench
        from coda_amd import CODA, Oracle, checkpoint as ckpt
        from coda_amd.datasets import Dataset
        from coda_amd.options import LOSS_FNS

        preds, labels = bench.synth_preds(list(range(12)), 700, 10, dev)
        ds = Dataset.from_tensors(preds, labels, dev)
        oracle = Oracle(ds, LOSS_FNS["acc"])

        def steps(sel, n, seed0):
            out = []
            for m in range(n):
                random.seed(seed0 + m)
                i, q = sel.get_next_item_to_label()
                sel.add_label(i, oracle(int(i)), q)
                out.append(int(i))
            return out

        random.seed(0); torch.manual_seed(0)
        ref = CODA(ds)
        t_ref = steps(ref, 8, 100)

        random.seed(0); torch.manual_seed(0)
        sel = CODA(ds)
        t_a = steps(sel, 4, 100)
        blob = ckpt.state_dict(sel)
        random.seed(0); torch.manual_seed(0)
        sel2 = CODA(ds)
        ckpt.load_state_dict(sel2, blob)
        t_b = steps(sel2, 4, 104)
        assert t_a + t_b == t_ref
        torch.testing.assert_close(sel2.get_pbest().cpu(),
                                   ref.get_pbest().cpu(),
                                   rtol=1e-4, atol=1e-6)


class TestSessionKernels:
    """Direct numerics for the late-round-2 kernels: each HIP op vs a
    plain fp32 torch reference on the same inputs."""

    def test_pi_marginal_two_stage(self):
        # deterministic two-stage reduce path (C % 4 == 0, C <= 1024)
        from coda_amd import ops
        assert ops.hip_available(), ops._ext_err
        torch.manual_seed(0)
        for (N, C) in [(50_000, 1000), (777, 128), (33, 4)]:
            adjusted = torch.rand(N, C, device="cuda") + 0.01
            row_sums = adjusted.sum(1) + 0.05
            out = ops._ext.pi_marginal(adjusted, row_sums)
            ref = (1.0 / row_sums.clamp_min(1e-12)) @ adjusted
            torch.testing.assert_close(out, ref, rtol=2e-5, atol=1e-4)
            # fixed-order schedule: bitwise repeatable
            out2 = ops._ext.pi_marginal(adjusted, row_sums)
            assert torch.equal(out, out2)

    def test_mixture_entropy_kernel(self):
        from coda_amd import ops
        from coda_amd.ops import reference
        torch.manual_seed(1)
        for (C, H) in [(1000, 128), (126, 8), (20, 16)]:
            rows = torch.rand(C, H, device="cuda")
            pi = torch.rand(C, device="cuda")
            pi = pi / pi.sum()
            m0, h0 = ops.mixture_entropy(rows, pi)
            rm, rh = reference.mixture_entropy(rows, pi)
            torch.testing.assert_close(m0, rm, rtol=1e-5, atol=1e-6)
            torch.testing.assert_close(h0.reshape(()), rh,
                                       rtol=1e-4, atol=1e-5)

    def test_pbest_single_row_vs_reference(self):
        from coda_amd import ops
        from coda_amd.ops import reference
        torch.manual_seed(2)
        for H in (3, 10, 128, 1000):
            a = torch.rand(1, H, device="cuda") * 50 + 0.5
            b = torch.rand(1, H, device="cuda") * 50 + 0.5
            k = ops._ext.pbest_from_beta(a, b, 256)
            ref = reference.pbest_from_beta(a.cpu(), b.cpu(), 256)
            torch.testing.assert_close(k.cpu(), ref,
                                       rtol=5e-3, atol=2e-4)

    def test_acq_select_vs_torch_chain(self):
        from coda_amd import ops
        torch.manual_seed(3)
        B = 10_000
        q0 = torch.randn(B, device="cuda")
        # exact duplicate maxima at known positions to exercise ties
        q0[[17, 444, 9999]] = q0.max() + 5.0
        h0 = torch.tensor([0.37], device="cuda")
        active = torch.rand(B, device="cuda") > 0.2
        active[[17, 444]] = True
        active[9999] = False   # masked-out duplicate must not count
        qbuf = torch.empty(B, device="cuda")
        out = torch.zeros(3, dtype=torch.float64, device="cuda")
        ties = torch.zeros(512, dtype=torch.int64, device="cuda")
        ops._ext.acq_select(q0, h0, active, qbuf, out, ties)
        q = torch.where(active, h0 + q0,
                        torch.full_like(q0, float("-inf")))
        bv, bi = q.max(0)
        nt = (torch.isclose(q, bv, rtol=1e-8) & active).sum()
        assert torch.equal(qbuf, q)
        o = out.cpu().tolist()
        assert o[0] == bv.item() and int(o[2]) == int(nt) == 2
        assert int(o[1]) in (17, 444) and int(o[1]) == min(17, 444)
        got = ties[1:1 + int(o[2])].cpu().tolist()
        assert sorted(v >> 32 for v in got) == [17, 444]

    def test_table_commit_row_vs_torch_chain(self):
        import bench
        from coda_amd import CODA
        from coda_amd.datasets import Dataset
        dev = "cuda:0"
        torch.manual_seed(4)
        preds, labels = bench.synth_preds(list(range(16)), 900, 12, dev)
        ds = Dataset.from_tensors(preds, labels, dev)
        sel = CODA(ds)                      # graphed: fused commit path
        i, q = sel.get_next_item_to_label()
        sel.add_label(i, 3, q)
        sel.get_next_item_to_label()        # forces the table refresh
        t = sel._tables
        # reproduce class 3's row with the torch chain
        from coda_amd import ops
        row = sel.dirichlets[:, 3, :]
        a_col = row[:, 3].contiguous()
        b_col = (row.sum(-1) - a_col).contiguous()
        eg, lc = ops._ext.beta_row_tables(a_col, b_col, 1.0)
        H = sel.Hl
        torch.testing.assert_close(t.EG[3].reshape(2 * H, -1),
                                   eg.reshape(2 * H, -1),
                                   rtol=0, atol=0)
        torch.testing.assert_close(t.delta[3], lc[:, 1] - lc[:, 0],
                                   rtol=0, atol=0)
        torch.testing.assert_close(t.s_base[3], lc[:, 0].sum(0),
                                   rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(
            t.egw[3].float(),
            (eg.reshape(2 * H, -1)
             * (torch.exp2(t.s_base[3]) * t.weights)).to(
                torch.bfloat16).float(),
            rtol=1e-2, atol=1e-3)

    def test_pi_hat_delta_class_major_mirror(self):
        from coda_amd import ops
        torch.manual_seed(5)
        for (H, N, C) in [(128, 20_000, 50), (64, 4_000, 33)]:
            preds = torch.rand(H, N, C, device="cuda")
            preds_t = preds.permute(0, 2, 1).contiguous()
            cls = torch.randint(0, C, (H,), device="cuda")
            a = ops.pi_hat_delta(preds, cls)
            b = ops.pi_hat_delta(preds, cls, preds_t=preds_t)
            # same accumulation pattern -> bitwise when the chunked
            # route is taken on both sides
            assert torch.equal(a, b)
            ref = preds[torch.arange(H, device="cuda"), :, cls].sum(0)
            torch.testing.assert_close(a, ref, rtol=1e-5, atol=1e-4)
