import random, sys, torch
sys.path.insert(0, ".")
import os
os.environ["CODA_AMD_ALLOW_EAGER"] = "1"
import coda_amd.ops as O
O._load_ext(); O._ext = None   # force eager everywhere on GPU
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
    q, cand = sel.eig_batched()
    ps, cls_rows = sel._pairs_static
    # per-candidate map: point id -> set of (c, frozenset hits)
    def pair_set(k):
        hs = set(ps.seg_h[int(ps.seg_off[k]):int(ps.seg_off[k+1])].tolist())
        if int(ps.pair_neg[k]):
            hs = set(range(8)) - hs
        return frozenset(hs)
    m = {}
    for row in range(ps.cand_ids.numel()):
        pid = int(ps.cand_ids[row])
        hits = set()
        for s in range(int(ps.cand_off[row]), int(ps.cand_off[row+1])):
            k = int(ps.cand_pairs[s])
            hits.add((int(ps.pair_c[k]), pair_set(k)))
        m[pid] = hits
    # h_after per (c,set) from the structure
    tables = sel._tables
    pb = sel._pbest_rows_before()
    mix, H0 = O.mixture_entropy(pb, sel.pi_hat)
    h = pops.pair_h_after(tables, ps, cls_rows, pb, sel.pi_hat, mix)
    hmap = {}
    for k in range(ps.K):
        if int(ps.pair_b[k]) < 0 and int(ps.seg_off[k+1]) == int(ps.seg_off[k]):
            continue
        hmap[(int(ps.pair_c[k]), pair_set(k))] = float(h[k])
    hbase = {c: float(h[int(ps.base_pos[c])]) for c in range(5)}
    return q.cpu(), list(cand), m, hmap, hbase, float(H0)

qs, cs, ms, hs, hb_s, H0s = probe(Comm(), dev)
r0 = run_ranks(4, lambda c: probe(c, dev), device=dev)[0]
qm, cm, mm, hm, hb_m, H0m = r0
print("H0 diff:", abs(H0s - H0m))
print("h_base diff:", max(abs(hb_s[c] - hb_m[c]) for c in range(5)))
bad_struct = [pid for pid in mm if mm[pid] != ms[pid]]
print("candidates with different hit structures:", len(bad_struct), bad_struct[:5])
common = set(hs) & set(hm)
wd = max(abs(hs[k] - hm[k]) for k in common)
print("h_after common keys:", len(common), "of", len(hm), "max diff:", wd)
only_m = set(hm) - set(hs)
print("sets only in sharded:", len(only_m))
