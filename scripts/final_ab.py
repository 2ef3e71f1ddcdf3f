"""Within-process A/B of the EIG engine configurations at bench scale.

Interleaved rounds (guide rule 24) on ONE box: fused kernel vs table
(fp32 GEMM) vs table (bf16 GEMM), full acquisition steps.
"""
import os, random, sys, time
sys.path.insert(0, ".")
import torch
from coda_amd import CODA, Oracle
from coda_amd.datasets import Dataset
from coda_amd.options import LOSS_FNS
import bench

dev = torch.device("cuda")
H, N, C = 128, 50_000, 1000
preds, labels = bench.synth_preds(list(range(H)), N, C, dev)
ds = Dataset.from_tensors(preds, labels, dev)
oracle = Oracle(ds, LOSS_FNS["acc"])

def make(impl, gemm):
    os.environ["CODA_AMD_V2_GEMM"] = gemm
    random.seed(0); torch.manual_seed(0)
    sel = CODA(ds, prefilter_n=256, chunk_size=256, eig_impl=impl)
    # warm: 3 steps (tables, hipblaslt algo cache)
    for _ in range(3):
        i, q = sel.get_next_item_to_label()
        sel.add_label(i, oracle(int(i)), q)
        sel.get_best_model_prediction()
    return sel

os.environ["CODA_AMD_NO_GRAPH"] = "1"
nograph = make("table", "bf16")
os.environ.pop("CODA_AMD_NO_GRAPH")
variants = {
    "fused": make("fused", "fp32"),
    "table-fp32": make("table", "fp32"),
    "table-bf16": make("table", "bf16"),
    "bf16-nograph": nograph,
}
res = {k: [] for k in variants}
for rnd in range(6):
    for name, sel in variants.items():
        os.environ["CODA_AMD_V2_GEMM"] = name.split("-")[-1] \
            if "-" in name else "fp32"
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            i, q = sel.get_next_item_to_label()
            sel.add_label(i, oracle(int(i)), q)
            sel.get_best_model_prediction()
        torch.cuda.synchronize()
        if rnd:
            res[name].append((time.perf_counter() - t0) / 5 * 1000)
lines = ["within-process interleaved A/B, 5-step rounds x5, ms/step "
         "(median | min), H=128 N=50k C=1000 prefilter=256:"]
for name, ts in res.items():
    ts.sort()
    lines.append(f"  {name:12s} {ts[len(ts)//2]:6.2f} | {ts[0]:6.2f}")
print("\n".join(lines), flush=True)
open("gpurun_out/engine_ab.txt", "w").write("\n".join(lines) + "\n")
