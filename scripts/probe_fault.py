import torch, sys
sys.path.insert(0, ".")
from coda_amd import ops
dev = torch.device("cuda")
def p(*a): print(*a, flush=True)

g = torch.Generator().manual_seed(0)

# bf16 bmm NaN bisection
for batch, N, C in [(2, 64, 8), (8, 1000, 100), (32, 50000, 1000), (16, 50000, 1000)]:
    pr = torch.rand(batch, N, C, device=dev)
    pr = pr / pr.sum(-1, keepdim=True)
    D = torch.rand(batch, C, C, device=dev)
    f32 = torch.bmm(pr, D.transpose(1, 2))
    b16 = torch.bmm(pr.to(torch.bfloat16), D.to(torch.bfloat16).transpose(1, 2))
    nan = int(b16.isnan().sum())
    err = (b16.float() - f32).abs().max() / f32.abs().max()
    p(f"bmm bf16 batch={batch} N={N} C={C}: nans={nan} relerr={float(err):.2e}")

# bf16 matmul non-batched
pr = torch.rand(50000, 1000, device=dev, dtype=torch.bfloat16)
D = torch.rand(1000, 1000, device=dev, dtype=torch.bfloat16)
r = pr @ D.t()
p("mm bf16 50000x1000x1000 nans:", int(r.isnan().sum()))

# non-contiguous transpose input?
D2 = torch.rand(32, 1000, 1000, device=dev, dtype=torch.bfloat16)
pr2 = torch.rand(32, 50000, 1000, device=dev, dtype=torch.bfloat16)
r2 = torch.bmm(pr2, D2.transpose(1, 2))
p("bmm bf16 transposed B: nans:", int(r2.isnan().sum()))
r3 = torch.bmm(pr2, D2.transpose(1, 2).contiguous())
p("bmm bf16 contig B: nans:", int(r3.isnan().sum()))
p("DONE")
