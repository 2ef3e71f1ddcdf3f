import random, sys, torch
sys.path.insert(0, ".")
import coda_amd.ops as O
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd import CODA
from coda_amd.ops import pair as pops

dev = "cuda:0"
H = 8
preds, labels = make_synthetic_task(H=H, N=200, C=5, seed=12)
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

def detail(ps, cls_rows, row, tag):
    A16 = O._ext.pair_dsum_es(tables.delta16, tables.dall, ps.pair_c,
                              ps.pair_neg, ps.seg_off, ps.seg_h)
    h = O._ext.pair_gemm_entropy(A16, tables.egw, ps.vmask, ps.pair_c,
                                 sel.pi_hat.contiguous(),
                                 pbest_before.contiguous(),
                                 mixture0.contiguous(), ps.tile)
    he = pops.pair_h_after(tables, ps, cls_rows, pbest_before,
                           sel.pi_hat, mixture0)
    pid = int(ps.cand_ids[row])
    print(f"--- {tag} candidate point {pid} (row {row}) tile={ps.tile} K={ps.K}")
    for s in range(int(ps.cand_off[row]), int(ps.cand_off[row + 1])):
        k = int(ps.cand_pairs[s])
        seg = ps.seg_h[int(ps.seg_off[k]):int(ps.seg_off[k+1])].tolist()
        print(f"  k={k} c={int(ps.pair_c[k])} neg={int(ps.pair_neg[k])} "
              f"seg={seg} h_kern={float(h[k]):.7f} h_eager={float(he[k]):.7f} "
              f"b_rep={int(ps.pair_b[k])}")
    hb = h[ps.base_pos]
    print("  h_base:", [f"{float(x):.7f}" for x in hb])

# full structure
mine_f = ids
cls_f = sel._global_classes(mine_f)
ps_f = pops.build_pairs(cls_f, mine_f, sel.C)
# slice 2
mine_s = ids[2::4]
cls_s = sel._global_classes(mine_s)
ps_s = pops.build_pairs(cls_s, mine_s, sel.C)
q_s = pops.eig_pairs(tables, ps_s, cls_s, pbest_before, sel.pi_hat,
                     mixture0, H_before, sel._adjusted, sel._row_sums)
d = (q_s.cpu() - q_full[2::4].cpu()).abs()
row_s = int(d.argmax())
pid = int(mine_s[row_s])
row_f = int((mine_f == pid).nonzero()[0])
print("worst pid", pid, "diff", float(d.max()),
      "q_slice", float(q_s[row_s]), "q_full", float(q_full[2::4][row_s]))
detail(ps_f, cls_f, row_f, "FULL")
detail(ps_s, cls_s, row_s, "SLICE")
