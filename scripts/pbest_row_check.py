"""Single-row pbest kernel: numerics vs eager + latency (GPU box)."""
import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from coda_amd import ops
assert ops.hip_available(), ops._ext_err
dev = "cuda:0"
torch.manual_seed(0)
for H in (3, 10, 128, 1000, 2048):
    a = (torch.rand(1, H, device=dev) * 50 + 0.5)
    b = (torch.rand(1, H, device=dev) * 50 + 0.5)
    k = ops._ext.pbest_from_beta(a, b, 256)
    ref = ops.reference.pbest_from_beta(a.cpu(), b.cpu(), 256)
    err = (k.cpu() - ref).abs().max().item()
    for _ in range(10):
        ops._ext.pbest_from_beta(a, b, 256)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100):
        ops._ext.pbest_from_beta(a, b, 256)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 100 * 1e6
    print(f"H={H:5d}  maxabs vs eager {err:.2e}  {us:7.1f} us")
