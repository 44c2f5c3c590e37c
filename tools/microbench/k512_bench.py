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
M, N, K = 716847, 512, 512  # yelp shape
g = torch.randn(M, N, device="cuda")
w1 = torch.randn(N, K, device="cuda")
w2 = torch.randn(N, K, device="cuda")
x1 = torch.randn(M, K, device="cuda")
x2 = torch.randn(M, K, device="cuda")
tf = 2 * 2 * M * N * K / 1e12
ms_f = t(lambda: nat.dual_dgrad(g, w1, w2))
ms_c = t(lambda: g @ torch.cat((w1, w2), dim=1))
print(f"dgrad K=512 yelp: fused {ms_f:.3f} ({tf/ms_f*1e3:.0f} TF) vs "
      f"rocBLAS cat {ms_c:.3f} ({tf/ms_c*1e3:.0f} TF)")
ms_w = t(lambda: nat.dual_wgrad(g, x1, x2))
ms_wr = t(lambda: (g.t() @ x1, g.t() @ x2))
print(f"wgrad K=512 yelp: fused {ms_w:.3f} ({tf/ms_w*1e3:.0f} TF) vs "
      f"rocBLAS pair {ms_wr:.3f} ({tf/ms_wr*1e3:.0f} TF)")
