import random, sys, torch
sys.path.insert(0, ".")
import coda_amd.ops as O
assert O.hip_available() or O._load_ext()
from coda_amd.datasets import Dataset, make_synthetic_task
from coda_amd.parallel import Comm
from coda_amd.ops import pair as pops, table as tops, reference as R

dev = "cuda:0"
preds, labels = make_synthetic_task(H=8, N=200, C=5, seed=12)
C = 5
cls = preds.argmax(-1).to(dev)   # (H, N)
g = torch.Generator().manual_seed(3)
dl = (torch.rand(8, C, C, generator=g) * 2 + 0.5).to(dev)
a, b = R.dirichlet_to_beta(dl)
tables = pops.attach_pair_tables(tops.table_precompute(a, b))
pb = R.pbest_from_beta(a.t().contiguous(), b.t().contiguous())
pi = torch.rand(C, generator=g).to(dev); pi /= pi.sum()
mix, H0 = R.mixture_entropy(pb, pi)

def h_after_for(ids):
    cr = cls[:, ids].t().to(torch.int32).contiguous()
    ps = pops.build_pairs(cr, ids, C)
    A16 = O._ext.pair_dsum_es(tables.delta16, tables.dall, ps.pair_c,
                              ps.pair_neg, ps.seg_off, ps.seg_h)
    h = O._ext.pair_gemm_entropy(A16, tables.egw, ps.vmask, ps.pair_c,
                                 pi.contiguous(), pb.contiguous(),
                                 mix.contiguous(), ps.tile)
    he = pops.pair_h_after(tables, ps, cr, pb, pi, mix)
    out = {}
    for k in range(ps.K):
        if int(ps.pair_b[k]) < 0 and int(ps.seg_off[k+1]) == int(ps.seg_off[k]):
            continue
        s = frozenset(ps.seg_h[int(ps.seg_off[k]):int(ps.seg_off[k+1])].tolist())
        if int(ps.pair_neg[k]):
            s = frozenset(range(8)) - s
        out[(int(ps.pair_c[k]), s)] = (float(h[k]), float(he[k]))
    return out

ids_all = torch.arange(200, device=dev)
full = h_after_for(ids_all)
sub = h_after_for(ids_all[0::4])
common = set(full) & set(sub)
worst = 0; worst_key = None
for key in common:
    d = abs(full[key][0] - sub[key][0])
    if d > worst: worst, worst_key = d, key
print("common pairs:", len(common), "kernel h_after worst diff:", worst, worst_key)
if worst_key:
    print("full (kern, eager):", full[worst_key], " sub:", sub[worst_key])
# eager-vs-kernel per run
for name, m in (("full", full), ("sub", sub)):
    dd = max(abs(k - e) for k, e in m.values())
    print(name, "kernel-vs-eager max:", dd)
