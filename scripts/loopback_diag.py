import random, sys, torch
sys.path.insert(0, ".")
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd.parallel.loopback import run_ranks
from coda_amd import CODA

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)

def q0(comm, device):
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    q, cand = sel.eig_batched()
    return q.cpu(), list(cand), sel.pi_hat.cpu(), sel._pbest_rows_before().cpu()

qs, cs, pis, rows_s = q0(Comm(), dev)
res = run_ranks(4, lambda comm: q0(comm, dev), device=dev)
qm, cm, pim, rows_m = res[0]
print("cand equal:", cs == cm, len(cs))
print("pi max diff:", float((pis - pim).abs().max()))
print("rows max diff:", float((rows_s - rows_m).abs().max()))
d = (qs - qm).abs()
print("q max diff:", float(d.max()), "at", int(d.argmax()), "of", len(cs))
bad = (d > 1e-5).sum()
print("n bad:", int(bad))
# which positions are bad - pattern by rank?
idx = torch.nonzero(d > 1e-5, as_tuple=True)[0]
print("bad positions mod 4:", torch.unique(idx % 4, return_counts=True))
print("single q[:12]:", qs[:12].tolist())
print("shard  q[:12]:", qm[:12].tolist())
# also CPU world-4 for reference
qsc, _, _, _ = q0(Comm(), "cpu")
resc = run_ranks(4, lambda comm: q0(comm, "cpu"))
print("cpu q max diff:", float((qsc - resc[0][0]).abs().max()))

# permutation check: are the sharded values the same multiset?
ss, _ = torch.sort(qs); sm, _ = torch.sort(qm)
print("sorted q max diff:", float((ss - sm).abs().max()))
# per-rank raw comparison: recompute rank0's slice directly
def q0_rank(comm, device):
    shard = (comm.rank, comm.world) if comm.world > 1 else None
    ds = Dataset.from_tensors(preds, labels, device, shard=shard)
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, comm=comm, eig_impl="pair", pi_hat_precision="fp32")
    cand = torch.tensor(list(sel._active_candidates), device=device)
    mine = cand[comm.rank::comm.world]
    return mine.cpu()
mines = run_ranks(4, lambda comm: q0_rank(comm, dev), device=dev)
cand_t = torch.tensor(cs)
for r in range(4):
    expect = cand_t[r::4]
    print(f"rank {r} ids match:", bool((mines[r] == expect).all()))
