import random, sys, torch
sys.path.insert(0, ".")
import coda_amd.ops as O
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd.parallel.loopback import run_ranks
from coda_amd import CODA

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)

def q0(comm, device, force_eager=False):
    if force_eager:
        O._ext = None  # after _load_ext, dispatch sees None + ALLOW_EAGER
        import os
        os.environ["CODA_AMD_ALLOW_EAGER"] = "1"
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    q, cand = sel.eig_batched()
    return q.cpu()

O._load_ext()
saved = O._ext
qs_k = q0(Comm(), dev)                       # single, kernels
qs_e = q0(Comm(), dev, force_eager=True)     # single, eager GPU
O._ext = saved
res = run_ranks(4, lambda c: q0(c, dev), device=dev)
qm_k = res[0]
O._ext = None
import os; os.environ["CODA_AMD_ALLOW_EAGER"] = "1"
res = run_ranks(4, lambda c: q0(c, dev), device=dev)
qm_e = res[0]
O._ext = saved
print("single kernel vs single eager :", float((qs_k - qs_e).abs().max()))
print("shard  eager  vs single eager :", float((qm_e - qs_e).abs().max()))
print("shard  kernel vs single kernel:", float((qm_k - qs_k).abs().max()))
print("shard  kernel vs shard eager  :", float((qm_k - qm_e).abs().max()))
