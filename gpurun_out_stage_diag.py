import torch
from pipegcn_amd import ops
for p in (0.1, 0.5):
    torch.manual_seed(1)
    x = torch.randn(4000, 130, device='cuda', requires_grad=True)
    y = ops.fused_dropout(x, p)
    kept = y != 0
    g = torch.randn_like(y)
    y.backward(g)
    a = x.grad[kept]; b = g[kept] / (1 - p)
    d = (a - b).abs(); rel = d / (b.abs() + 1e-12)
    bad = rel > 1e-3
    print(p, 'maxrel', rel.max().item(), 'nbad', bad.sum().item(),
          'dropped-nonzero-grad', (x.grad[~kept] != 0).sum().item(),
          'frac kept', kept.float().mean().item())
    if bad.any():
        print('  a', a[bad][:6].tolist())
        print('  b', b[bad][:6].tolist())
