import random, sys, torch
sys.path.insert(0, ".")
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd.parallel.loopback import run_ranks
from coda_amd import CODA
import coda_amd.ops as O

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)

def probe(comm, device):
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    classes, ens_sum = O.init_model_stats(ds.preds)
    comm.all_reduce_(ens_sum)
    pseudo = (ens_sum / 8).argmax(-1)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    return (pseudo.cpu(), ens_sum.cpu(), sel.dirichlets.cpu(),
            sel._beta_view()[0].cpu())

ps_s = probe(Comm(), dev)
ps_m = run_ranks(4, lambda c: probe(c, dev), device=dev)[0]
print("pseudo flips:", int((ps_s[0] != ps_m[0]).sum()))
print("ens max diff:", float((ps_s[1] - ps_m[1]).abs().max()))
d = (ps_s[2][0:8:4] if False else None)
print("alpha_g max diff vs single alpha:",
      float((ps_s[3] - ps_m[3]).abs().max()))
