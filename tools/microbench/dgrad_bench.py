import sys, time
sys.path.insert(0, ".")
import torch
from pipegcn_amd import native

def t(f, n=30):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3

nat = native()
for (M, N, K) in ((232965, 256, 602), (232965, 256, 256)):
    g = torch.randn(M, N, device="cuda")
    w1 = torch.randn(N, K, device="cuda")
    w2 = torch.randn(N, K, device="cuda")
    ms_f = t(lambda: nat.dual_dgrad(g, w1, w2))
    ms_c = t(lambda: g @ torch.cat((w1, w2), dim=1))
    tf = 2 * 2 * M * N * K / 1e12
    print(f"M{M} N{N} K{K}: dgrad fused {ms_f:.3f} ms ({tf/ms_f*1e3:.0f} TF)"
          f" vs rocBLAS cat {ms_c:.3f} ms ({tf/ms_c*1e3:.0f} TF)", flush=True)

