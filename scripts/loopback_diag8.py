import random, sys, torch
sys.path.insert(0, ".")
import coda_amd.ops as O
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
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
for r in range(4):
    mine = ids[r::4]
    cls_rows = sel._global_classes(mine)
    ps = pops.build_pairs(cls_rows, mine, sel.C)
    q_slice = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                             sel.pi_hat, mixture0, H_before,
                             sel._adjusted, sel._row_sums)
    print("rank-slice", r, "max diff vs full:",
          float((q_slice.cpu() - q_full[r::4].cpu()).abs().max()))
