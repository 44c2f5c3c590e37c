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
M, K, N = 232965, 256, 41
x1 = torch.randn(M, K, device="cuda")
x2 = torch.randn(M, K, device="cuda")
w1 = torch.randn(N, K, device="cuda")
w2 = torch.randn(N, K, device="cuda")
b = torch.randn(N, device="cuda")
def roc():
    out = torch.mm(x1, w1.t())
    out.addmm_(x2, w2.t())
    out.add_(b)
    return out
ms_r = t(roc)
ms_k = t(lambda: nat.sage_dual_gemm(x1, x2, w1, w2, b))
ref = roc()
out = nat.sage_dual_gemm(x1, x2, w1, w2, b)
err = (out - ref).abs().max().item() / ref.abs().max().item()
print(f"thin fwd N=41: rocBLAS {ms_r:.3f} ms vs MFMA kernel {ms_k:.3f} ms"
      f" (rel err {err:.2e})")
# dropout after vectorization
x = torch.randn(M, 256, device="cuda")
from pipegcn_amd import ops
ms_d = t(lambda: ops.fused_dropout(x, 0.5))
print(f"dropout fwd [233k,256]: {ms_d:.3f} ms")
