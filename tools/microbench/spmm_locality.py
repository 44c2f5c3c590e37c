"""SpMM gather rate vs graph locality (evidence for profiles/README.md):
the synthetic bench graph draws sources uniformly (no locality to
exploit); real graphs (reddit/papers after their release orderings)
cluster sources near destinations. Measures the SAME kernel on both."""
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

n, deg = 232965, 492
g = torch.Generator().manual_seed(0)
dst = torch.repeat_interleave(torch.arange(n), deg)
E = dst.numel()

def bench(name, src):
    csr = CSR.from_coo(src, dst, n, n).to("cuda")
    for F in (602, 256):
        x = torch.randn(n, F, device="cuda")
        ms_lpt = t(lambda: ops.spmm(csr, x, None))
        ro = csr.row_order
        csr.row_order = None
        ms_nat = t(lambda: ops.spmm(csr, x, None))
        csr.row_order = ro
        gb = E * F * 4 / 1e9
        print(f"{name} F={F}: LPT {ms_lpt:.1f} ms ({gb/ms_lpt:.1f} TB/s "
              f"logical), natural-order {ms_nat:.1f} ms "
              f"({gb/ms_nat:.1f} TB/s)", flush=True)

# uniform sources (the bench graph's structure at world-1)
bench("uniform ", (torch.rand(E, generator=g) * n).long())
# power-law locality: src clustered around dst (real-graph-like)
mag = (torch.rand(E, generator=g).clamp(min=1e-9) ** -2.0).long() + 1
sign = torch.where(torch.rand(E, generator=g) < 0.5, -1, 1)
bench("locality", (dst + sign * mag).remainder(n))
