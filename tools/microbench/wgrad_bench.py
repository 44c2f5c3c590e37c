import sys, time
sys.path.insert(0, ".")
import torch
from pipegcn_amd import native

def t(f, n=20):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3

nat = native()
for (M, N, K) in ((232965, 256, 602), (232965, 256, 256),
                  (232965, 41, 256), (2449029, 256, 256)):
    g = torch.randn(M, N, device="cuda")
    x1 = torch.randn(M, K, device="cuda")
    x2 = torch.randn_like(x1)
    ms_f = t(lambda: nat.dual_wgrad(g, x1, x2))
    ms_r = t(lambda: (g.t() @ x1, g.t() @ x2))
    w = torch.randn(N, K, device="cuda")
    ms_dc = t(lambda: g @ torch.cat((w, w), dim=1))
    ms_dr = t(lambda: (g @ w, g @ w))
    tf = 2 * 2 * M * N * K / 1e12
    print(f"M{M} N{N} K{K}: wgrad fused {ms_f:.3f} ms ({tf/ms_f*1e3:.0f} TF)"
          f" vs rocBLAS pair {ms_r:.3f} ms ({tf/ms_r*1e3:.0f} TF) | "
          f"dgrad cat {ms_dc:.3f} vs pair {ms_dr:.3f}", flush=True)
