"""Multi-rank simulation: the sharded paths at world 3/4 on one device.

Complements the 2-rank gloo subprocess tests: here the SAME process runs
W threads through the real sharded code (including, on GPU, the sharded
table kernels at world > 1 - which a 1-GPU box cannot otherwise test).
"""
import random

import pytest
import torch

from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel.loopback import run_ranks


def _trajectory(comm, preds, labels, device, steps=3, **kw):
    from coda_amd import CODA, Oracle
    from coda_amd.options import LOSS_FNS
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    oracle = Oracle(ds, LOSS_FNS["acc"])
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, chunk_size=32, **kw)
    traj = []
    for _ in range(steps):
        idx, q = sel.get_next_item_to_label()
        sel.add_label(idx, oracle(int(idx)), q)
        traj.append((int(idx), round(float(q), 5)))
    return traj, sel.get_pbest().cpu()


@pytest.mark.parametrize("world", [3, 4])
def test_loopback_sharded_equals_single_cpu(world):
    preds, labels = make_synthetic_task(H=7, N=150, C=4, seed=9)
    from coda_amd.parallel import Comm
    single_traj, single_pbest = _trajectory(Comm(), preds, labels, "cpu")

    results = run_ranks(
        world, lambda comm: _trajectory(comm, preds, labels, "cpu"))
    for traj, pbest in results:
        assert traj == single_traj
        torch.testing.assert_close(pbest, single_pbest, rtol=1e-4,
                                   atol=1e-6)


@pytest.mark.parametrize("impl", ["fused", "table"])
def test_loopback_both_engines_cpu(impl):
    preds, labels = make_synthetic_task(H=6, N=120, C=4, seed=10)
    from coda_amd.parallel import Comm
    st, sp = _trajectory(Comm(), preds, labels, "cpu", eig_impl=impl)
    results = run_ranks(
        3, lambda comm: _trajectory(comm, preds, labels, "cpu",
                                    eig_impl=impl))
    for traj, pbest in results:
        assert traj == st
        torch.testing.assert_close(pbest, sp, rtol=1e-4, atol=1e-6)


@pytest.mark.gpu
def test_loopback_sharded_table_gpu():
    """World-4 sharded TABLE path (es_build_gathered / eig_totals /
    eig_entropy kernels + gathers) on one GPU vs single-device."""
    assert torch.cuda.is_available()
    dev = "cuda:0"
    preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=11)
    from coda_amd.parallel import Comm
    st, sp = _trajectory(Comm(), preds, labels, dev, eig_impl="table")
    results = run_ranks(
        4, lambda comm: _trajectory(comm, preds, labels, dev,
                                    eig_impl="table"), device=dev)
    for traj, pbest in results:
        assert traj == st, (traj, st)
        torch.testing.assert_close(pbest, sp, rtol=2e-3, atol=1e-5)


@pytest.mark.gpu
def test_loopback_replicated_pair_gpu():
    """World-4 replicated-beta candidate-sharded PAIR path (the v3
    default for sharded runs: init gathers + pair kernels + the (B,)
    EIG gather) on one GPU vs single-device."""
    assert torch.cuda.is_available()
    dev = "cuda:0"
    preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)
    from coda_amd.parallel import Comm
    # fp32 pi_hat isolates the sharding logic: the bf16 pi fast path
    # rounds differently per GEMM shape (documented deviation), which
    # would otherwise add ~1e-3-relative noise on top of sharding
    st, sp = _trajectory(Comm(), preds, labels, dev, eig_impl="pair",
                         pi_hat_precision="fp32")
    results = run_ranks(
        4, lambda comm: _trajectory(comm, preds, labels, dev,
                                    eig_impl="pair",
                                    pi_hat_precision="fp32"),
        device=dev)
    for traj, pbest in results:
        assert [t[0] for t in traj] == [t[0] for t in st], (traj, st)
        torch.testing.assert_close(pbest, sp, rtol=2e-3, atol=1e-5)
        for (_, qa), (_, qb) in zip(traj, st):
            assert abs(qa - qb) <= 1e-3 * max(1.0, abs(qb))
