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

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)

def probe(comm, device):
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    q, cand = sel.eig_batched()
    return (q.cpu(), sel._adjusted.cpu(), sel._row_sums.cpu(),
            sel.pi_hat.cpu())

qs, adj_s, rs_s, pi_s = probe(Comm(), dev)
qm, adj_m, rs_m, pi_m = run_ranks(4, lambda c: probe(c, dev),
                                  device=dev)[0]
print("adjusted max diff:", float((adj_s - adj_m).abs().max()),
      "rel:", float(((adj_s - adj_m).abs() / adj_s.abs().clamp_min(1e-9)).max()))
print("row_sums max diff:", float((rs_s - rs_m).abs().max()))
print("q max diff:", float((qs - qm).abs().max()))
b = int((qs - qm).abs().argmax())
print("worst b:", b, "adj row diff:", float((adj_s[b+0] - adj_m[b+0]).abs().max()) if b < 200 else None)
