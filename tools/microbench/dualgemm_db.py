import os, sys, time
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
for (M, K, N) in ((232965, 602, 256), (232965, 256, 256)):
    x1 = torch.randn(M, K, device="cuda")
    x2 = torch.randn(M, K, device="cuda")
    w1 = torch.randn(N, K, device="cuda")
    w2 = torch.randn(N, K, device="cuda")
    b = torch.randn(N, device="cuda")
    ref = (x1 @ w1.t()) + (x2 @ w2.t()) + b
    out = nat.sage_dual_gemm(x1, x2, w1, w2, b)
    err = (out - ref).abs().max().item() / ref.abs().max().item()
    ms = t(lambda: nat.sage_dual_gemm(x1, x2, w1, w2, b))
    tf = 2 * 2 * M * N * K / 1e12
    mode = "DB" if os.environ.get("PIPEGCN_GEMM_DB") == "1" else "SB"
    print(f"{mode} M{M} K{K} N{N}: {ms:.3f} ms ({tf/ms*1e3:.0f} TF) "
          f"rel err {err:.2e}", flush=True)
