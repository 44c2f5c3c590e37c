import sys, time
sys.path.insert(0, ".")
import torch
from pipegcn_amd import ops

def t(f, n=30):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3

ln = torch.nn.LayerNorm(256).cuda()
x = torch.randn(232965, 256, device="cuda", requires_grad=True)
out = ops.layer_norm_relu(x, ln, relu=True)
ref = torch.nn.functional.relu(torch.nn.functional.layer_norm(
    x, (256,), ln.weight, ln.bias, ln.eps))
print("fwd maxerr:", (out - ref).abs().max().item())
g = torch.randn_like(out)
ms_f = t(lambda: ops.layer_norm_relu(x, ln, relu=True))
def bwd():
    x.grad = None
    o = ops.layer_norm_relu(x, ln, relu=True)
    o.backward(g)
ms_fb = t(bwd)
print(f"ln fwd {ms_f:.3f} ms, fwd+bwd {ms_fb:.3f} ms")
# correctness of backward
x.grad = None
ln.weight.grad = ln.bias.grad = None
ops.layer_norm_relu(x, ln, relu=True).backward(g)
gx = x.grad.clone()
x.grad = None
torch.nn.functional.relu(torch.nn.functional.layer_norm(
    x, (256,), ln.weight, ln.bias, ln.eps)).backward(g)
print("bwd maxerr:", (gx - x.grad).abs().max().item())
