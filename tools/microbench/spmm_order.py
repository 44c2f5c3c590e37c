"""Row-order policy on a REAL-like graph (locality + lognormal degrees):
global LPT (degree sort) balances waves but scatters the near-diagonal
source locality; banded LPT sorts by degree within contiguous row bands,
keeping both."""
import sys, time
sys.path.insert(0, ".")
import torch
from pipegcn_amd import ops
from pipegcn_amd.graph.csr import CSR

def t(f, n=10):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3

n, avg = 232965, 492
g = torch.Generator().manual_seed(0)
d = torch.exp(torch.randn(n, generator=g))
deg = (d / d.mean() * avg).long().clamp(min=1)
dst = torch.repeat_interleave(torch.arange(n), deg)
E = dst.numel()
mag = (torch.rand(E, generator=g).clamp(min=1e-9) ** -2.0).long() + 1
sign = torch.where(torch.rand(E, generator=g) < 0.5, -1, 1)
src = (dst + sign * mag).remainder(n)
csr = CSR.from_coo(src, dst, n, n).to("cuda")
rdeg = (csr.indptr[1:] - csr.indptr[:-1]).cpu()

orders = {"natural": None,
          "global LPT": csr.row_order}
band = 16384
bo = []
for s in range(0, n, band):
    e = min(n, s + band)
    idx = torch.argsort(rdeg[s:e], descending=True) + s
    bo.append(idx)
orders["banded LPT(16k)"] = torch.cat(bo).to(torch.int32).cuda()

for F in (602, 256):
    x = torch.randn(n, F, device="cuda")
    line = f"F={F} ({E} edges):"
    for name, ro in orders.items():
        csr.row_order = ro
        ms = t(lambda: ops.spmm(csr, x, None))
        line += f"  {name} {ms:.1f} ms"
    print(line, flush=True)
