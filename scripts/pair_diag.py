"""Diagnose pair-kernel numerics vs the eager pair math (GPU)."""
import random
import sys

import torch

sys.path.insert(0, ".")
import coda_amd.ops as O
from coda_amd.ops import pair as pops
from coda_amd.ops import table as tops
from coda_amd.ops import reference as R
from tests.test_pair import _random_problem

assert O.hip_available(), "extension missing"
dev = torch.device("cuda:0")

for H, N, C in [(10, 120, 7), (3, 60, 2), (16, 200, 126), (128, 500, 50),
                (128, 6000, 1000)]:
    (preds, cls, dirichlets, pi_hat, adjusted,
     row_sums) = _random_problem(H, N, C, seed=H + C)
    alpha_cc, beta_cc = R.dirichlet_to_beta(dirichlets)
    alpha_cc, beta_cc = alpha_cc.to(dev), beta_cc.to(dev)
    tables = tops.table_precompute(alpha_cc, beta_cc)
    tables = pops.attach_pair_tables(tables)
    pbest_before = R.pbest_from_beta(alpha_cc.t().contiguous(),
                                     beta_cc.t().contiguous())
    mixture0, H_before = R.mixture_entropy(pbest_before, pi_hat.to(dev))
    ids = torch.arange(N, device=dev)
    cls_rows = cls.to(dev)[:, ids].t().to(torch.int32).contiguous()
    ps = pops.build_pairs(cls_rows, ids, C)
    eig_k = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                           pi_hat.to(dev), mixture0, H_before,
                           adjusted.to(dev), row_sums.to(dev))
    h_e = pops.pair_h_after(tables, ps, cls_rows, pbest_before,
                            pi_hat.to(dev), mixture0)
    h_k = O._ext.pair_gemm_entropy(
        O._ext.pair_dsum_es(tables.delta16, tables.dall, ps.pair_c,
                            ps.pair_neg, ps.seg_off, ps.seg_h),
        tables.egw, ps.vmask, ps.pair_c,
        pi_hat.to(dev).contiguous(), pbest_before.contiguous(),
        mixture0.contiguous(), ps.tile)
    eig_e = pops.eig_from_pairs(h_e, ps, adjusted.to(dev),
                                row_sums.to(dev), H_before)[ps.cand_ids]
    dh = (h_k - h_e).abs()
    de = (eig_k - eig_e).abs()
    scale = eig_e.abs().max()
    print(f"H={H} N={N} C={C} K={ps.K} tile={ps.tile}: "
          f"|dh| max={dh.max():.3e}  |deig| max={de.max():.3e} "
          f"mean={de.mean():.3e}  eig_scale={scale:.3e} "
          f"rel={float(de.max()/scale):.3e} "
          f"nonfinite={int((~torch.isfinite(h_k)).sum())}")

# trajectory divergence probe
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
    traj, qtop = [], []
    for _ in range(8):
        q_vals, cand = sel.eig_batched()
        tv, ti = q_vals.topk(min(5, q_vals.numel()))
        qtop.append((tv.tolist(), ti.tolist()))
        i, q = sel.get_next_item_to_label()
        sel.add_label(i, oracle(int(i)), q)
        traj.append(int(i))
    return traj, qtop


t_t, q_t = run("table")
t_p, q_p = run("pair")
print("table traj", t_t)
print("pair  traj", t_p)
for s, (a, b) in enumerate(zip(q_t, q_p)):
    flag = "DIVERGE" if t_t[s] != t_p[s] else ""
    print(f"step {s} {flag}\n  table top {[f'{v:.6e}' for v in a[0]]} {a[1]}"
          f"\n  pair  top {[f'{v:.6e}' for v in b[0]]} {b[1]}")
    if flag:
        break

# add_label path probe: is the hipGraph live in the bench configuration?
preds, labels = bench.synth_preds(list(range(8)), 2000, 40, dev)
ds2 = Dataset.from_tensors(preds, labels, dev)
oracle2 = Oracle(ds2, LOSS_FNS["acc"])
random.seed(0); torch.manual_seed(0)
sel = CODA(ds2)
for _ in range(4):
    i, q = sel.get_next_item_to_label()
    sel.add_label(i, oracle2(int(i)), q)
    sel.get_best_model_prediction()
print("label graph live:", sel._label_graph is not None,
      "impl pair static:", sel._pairs_static is not None)
