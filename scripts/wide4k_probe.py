import random, sys, time
sys.path.insert(0, ".")
import torch, bench
from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset
from coda_amd.options import LOSS_FNS
dev = torch.device("cuda")
H, N, C = 4096, 5000, 100
preds, labels = bench.synth_preds(list(range(H)), N, C, dev)
ds = Dataset.from_tensors(preds.to(torch.bfloat16), labels, dev)
ds.total_models = H
oracle = Oracle(ds, LOSS_FNS["acc"])
random.seed(0); torch.manual_seed(0)
t0 = time.perf_counter()
sel = CODA(ds, prefilter_n=128, chunk_size=128)
torch.cuda.synchronize()
print(f"init (H=4096, wide-H fallback for class rows): {time.perf_counter()-t0:.2f}s", flush=True)
for s in range(3):
    t0 = time.perf_counter()
    i, q = sel.get_next_item_to_label()
    sel.add_label(i, oracle(int(i)), q)
    best = sel.get_best_model_prediction()
    torch.cuda.synchronize()
    print(f"step {s}: {1000*(time.perf_counter()-t0):.1f} ms, best={int(best)}", flush=True)
p = sel.get_pbest()
assert p.shape == (H,) and torch.isfinite(p).all()
print("WIDE4K OK")
