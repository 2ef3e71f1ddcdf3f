"""Probe: bf16 vs f32 for the v2 (C,B,P)x(P,2H) batched GEMM (GPU box).

Checks speed + NaN/fault behavior at the exact v2 shapes (batch=1000 of
256x256x256) and the EIG accuracy impact of bf16 ES/EG inputs.
"""
import sys, time
sys.path.insert(0, ".")
import torch
from coda_amd import ops
from coda_amd.ops import table as T

dev = torch.device("cuda")
H, C, B, P = 128, 1000, 256, 256
g = torch.Generator().manual_seed(0)
a0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
b0 = (torch.rand(H, C, generator=g) * 20 + 1).to(dev)
cls = torch.randint(0, C, (B, H), generator=g).to(dev)
pi = torch.softmax(torch.rand(C, generator=g), 0).to(dev)
pixi = torch.softmax(torch.rand(B, C, generator=g), -1).to(dev)
pb0 = ops.pbest_from_beta(a0.t().contiguous(), b0.t().contiguous())
m0, H0 = ops.mixture_entropy(pb0, pi)
tables = T.table_precompute(a0, b0)
EG, delta, s_base, w = tables
hv, off = T._class_csr(cls.long(), C)
cls32 = cls.to(torch.int32)
ES = ops._ext.es_build(s_base, delta, hv, off, w)
EGr = EG.reshape(C, 2 * H, P)

def t(fn, reps=15):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1000

M32 = torch.bmm(ES, EGr.transpose(1, 2))
print(f"f32 bmm: {t(lambda: torch.bmm(ES, EGr.transpose(1,2))):.3f} ms", flush=True)

ES16 = ES.to(torch.bfloat16)
EG16 = EGr.to(torch.bfloat16)
M16 = torch.bmm(ES16, EG16.transpose(1, 2))
print(f"bf16 bmm: {t(lambda: torch.bmm(ES16, EG16.transpose(1,2))):.3f} ms", flush=True)
print(f"bf16 nans: {int(M16.isnan().sum())} infs: {int(M16.isinf().sum())}", flush=True)

# EIG impact
ha32 = ops._ext.eig_assemble_k(M32, cls32, pi, pb0, m0)
ha16 = ops._ext.eig_assemble_k(M16.float(), cls32, pi, pb0, m0)
e32 = H0 - (pixi * ha32).sum(-1)
e16 = H0 - (pixi * ha16).sum(-1)
d = (e32 - e16).abs()
print(f"EIG abs diff: max {float(d.max()):.3e} (EIG scale ~{float(e32.abs().max()):.2e})")
print(f"argmax agree: {int(e32.argmax()) == int(e16.argmax())}")
print("BF16 PROBE DONE")
