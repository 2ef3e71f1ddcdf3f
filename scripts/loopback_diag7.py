import random, sys, torch
sys.path.insert(0, ".")
import os
os.environ["CODA_AMD_ALLOW_EAGER"] = "1"
import coda_amd.ops as O
O._load_ext(); O._ext = None
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd.parallel.loopback import run_ranks
from coda_amd import CODA
from coda_amd.ops import pair as pops

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)

def probe(comm, device):
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    # raw local q BEFORE the gather: recompute exactly what _eig_pair does
    from coda_amd import ops as OO
    alpha_cc, beta_cc = sel._beta_view()
    tables = sel._refresh_tables(alpha_cc, beta_cc, want_egw=True)
    pbest_before = sel._pbest_rows_before()
    mixture0, H_before = OO.mixture_entropy(pbest_before, sel.pi_hat)
    ids = torch.tensor(list(sel._active_candidates), device=device)
    mine = ids[comm.rank::comm.world] if comm.world > 1 else ids
    cls_rows = sel._global_classes(mine)
    ps = pops.build_pairs(cls_rows, mine, sel.C)
    q_local = pops.eig_pairs(tables, ps, cls_rows, pbest_before,
                             sel.pi_hat, mixture0, H_before,
                             sel._adjusted, sel._row_sums)
    q_full, cand = sel.eig_batched()
    return (q_local.cpu(), mine.cpu(), q_full.cpu(), float(H_before))

qs_loc, ids_s, qs_full, H0s = probe(Comm(), dev)
res = run_ranks(4, lambda c: probe(c, dev), device=dev)
print("H_before per rank:", [r[3] for r in res], "single:", H0s)
# reassemble by hand from raw locals
n = ids_s.numel()
qhand = torch.empty(n)
for r in range(4):
    qhand[r::4] = res[r][0]
print("hand-reassembled vs single:", float((qhand - qs_full).abs().max()))
print("hand vs sharded-returned:", float((qhand - res[0][2]).abs().max()))
for r in range(4):
    mine = ids_s[r::4]
    print(f"rank{r} ids ok:", bool((res[r][1] == mine).all()),
          "local vs single-slice:",
          float((res[r][0] - qs_full[r::4]).abs().max()))
