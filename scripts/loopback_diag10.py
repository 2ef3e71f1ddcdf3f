import random, sys, torch
sys.path.insert(0, ".")
import coda_amd.ops as O
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd import CODA
from coda_amd.ops import pair as pops

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)
ds = Dataset.from_tensors(preds, labels, dev)
random.seed(0); torch.manual_seed(0)
sel = CODA(ds, eig_impl="pair", pi_hat_precision="fp32")
q_full, cand = sel.eig_batched()

from coda_amd import ops as OO
alpha_cc, beta_cc = sel._beta_view()
tables = sel._refresh_tables(alpha_cc, beta_cc, want_egw=True)
pbest_before = sel._pbest_rows_before()
mixture0, H_before = OO.mixture_entropy(pbest_before, sel.pi_hat)
ids = torch.tensor(list(sel._active_candidates), device=dev)

def manual_q(ps, pid):
    A16 = O._ext.pair_dsum_es(tables.delta16, tables.dall, ps.pair_c,
                              ps.pair_neg, ps.seg_off, ps.seg_h)
    h = O._ext.pair_gemm_entropy(A16, tables.egw, ps.vmask, ps.pair_c,
                                 sel.pi_hat.contiguous(),
                                 pbest_before.contiguous(),
                                 mixture0.contiguous(), ps.tile).double()
    hb = h[ps.base_pos]
    row = int((ps.cand_ids == pid).nonzero()[0])
    arow = sel._adjusted[pid].double()
    inv = 1.0 / max(float(sel._row_sums[pid]), 1e-12)
    base = float(arow @ hb)
    corr = 0.0
    for s in range(int(ps.cand_off[row]), int(ps.cand_off[row + 1])):
        k = int(ps.cand_pairs[s])
        c = int(ps.pair_c[k])
        corr += float(arow[c]) * (float(h[k]) - float(hb[c]))
    return float(H_before) - (base + corr) * inv

pid = 118
ps_f = pops.build_pairs(sel._global_classes(ids), ids, sel.C)
mine_s = ids[2::4]
ps_s = pops.build_pairs(sel._global_classes(mine_s), mine_s, sel.C)
qm_f = manual_q(ps_f, pid)
qm_s = manual_q(ps_s, pid)
row_f = int((ids == pid).nonzero()[0])
print("manual q (full struct):", qm_f)
print("manual q (slice struct):", qm_s)
print("returned q_full:", float(q_full[row_f]))
q_s = pops.eig_pairs(tables, ps_s, sel._global_classes(mine_s),
                     pbest_before, sel.pi_hat, mixture0, H_before,
                     sel._adjusted, sel._row_sums)
row_s = int((mine_s == pid).nonzero()[0])
print("returned q_slice:", float(q_s[row_s]))
