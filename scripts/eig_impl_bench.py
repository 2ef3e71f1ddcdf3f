"""A/B the fused-kernel vs table-factored EIG at bench scale (GPU box)."""
import os, sys, time
sys.path.insert(0, ".")
import torch

os.environ.setdefault("CODA_BENCH_H", "128")
from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset
from coda_amd.options import LOSS_FNS
import bench, random

dev = torch.device("cuda")
H, N, C = 128, 50_000, 1000
preds, labels = bench.synth_preds(list(range(H)), N, C, dev)
ds = Dataset.from_tensors(preds, labels, dev)
oracle = Oracle(ds, LOSS_FNS["acc"])

res = {}
for impl in ("fused", "table"):
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, prefilter_n=256, chunk_size=256, eig_impl=impl)
    e, c = sel.eig_batched()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        e, c = sel.eig_batched()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    res[impl] = (dt, e.clone())
    print(f"{impl:6s}: eig_batched {1000*dt:8.2f} ms", flush=True)
d = (res["fused"][1] - res["table"][1]).abs()
rel = d / res["fused"][1].abs().clamp_min(1e-9)
print(f"agreement: max abs diff {float(d.max()):.3e}, "
      f"max rel {float(rel.max()):.3e}")
a1 = int(res["fused"][1].argmax()); a2 = int(res["table"][1].argmax())
print("argmax agree:", a1 == a2, a1, a2)
