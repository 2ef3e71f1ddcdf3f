"""Host-side split of one bench step (GPU): where the 1.4 ms goes."""
import random, sys, time, torch
sys.path.insert(0, ".")
import bench
from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset
from coda_amd.options import LOSS_FNS

dev = "cuda:0"
preds, labels = bench.synth_preds(list(range(128)), 50_000, 1000, dev)
ds = Dataset.from_tensors(preds, labels, dev)
oracle = Oracle(ds, LOSS_FNS["acc"])
random.seed(0); torch.manual_seed(0)
sel = CODA(ds)
for _ in range(30):
    i, q = sel.get_next_item_to_label()
    sel.add_label(i, oracle(int(i)), q)
    sel.get_best_model_prediction()
torch.cuda.synchronize()

N = 300
t_next = t_or = t_lab = t_best = 0.0
t0 = time.perf_counter()
for _ in range(N):
    a = time.perf_counter()
    i, q = sel.get_next_item_to_label()
    b = time.perf_counter()
    y = oracle(int(i))
    c = time.perf_counter()
    sel.add_label(i, y, q)
    d = time.perf_counter()
    sel.get_best_model_prediction()
    e = time.perf_counter()
    t_next += b - a; t_or += c - b; t_lab += d - c; t_best += e - d
torch.cuda.synchronize()
wall = time.perf_counter() - t0
print(f"wall/step {1000*wall/N:.3f} ms | get_next {1000*t_next/N:.3f} "
      f"oracle {1000*t_or/N:.3f} add_label {1000*t_lab/N:.3f} "
      f"best {1000*t_best/N:.3f}")
