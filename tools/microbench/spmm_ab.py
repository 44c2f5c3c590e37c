import sys, time
sys.path.insert(0, ".")
import torch
from pipegcn_amd import ops
from pipegcn_amd.graph.synthetic import synth_partition
from pipegcn_amd.graph.halo import build_runtime_partition

dev = "cuda" if torch.cuda.is_available() else "cpu"
part = synth_partition("reddit" if dev == "cuda" else "small", 0, 1,
                       seed=0, train_frac=0.66)
rp = build_runtime_partition(part, device=dev)
inv = (1.0 / rp.ndata["in_degree"].clamp(min=1.0)).contiguous()

def t(f, n=20):
    for _ in range(3):
        f()
    if dev == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    if dev == "cuda":
        torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3

for F in (602, 256):
    x = torch.randn(rp.num_all, F, device=dev)
    g = torch.randn(rp.num_in, F, device=dev)
    ms = t(lambda: ops.spmm(rp.graph.csr, x, inv))
    msb = t(lambda: ops.spmm(rp.graph.csc, g, None, src_scale=inv))
    print(f"F={F}: fwd {ms:.2f} ms, bwd(csc) {msb:.2f} ms", flush=True)
