import sys

sys.path.insert(0, ".")
import torch  # noqa: E402

from pipegcn_amd import native, ops  # noqa: E402

n = native()
x = torch.randn(520000, device="cuda")
y, mask = n.dropout_fwd(x, 0.5, 1234)
bits = ((mask.unsqueeze(1) >>
         torch.arange(8, device="cuda", dtype=torch.uint8)) &
        1).flatten()[: x.numel()].bool()
print("fwd kept==bits:", torch.equal(y != 0, bits))
print("fwd scale maxrel:",
      ((y[bits] - x[bits] * 2).abs() /
       (x[bits].abs() * 2 + 1e-12)).max().item())
dy = torch.randn_like(x)
dx = n.dropout_bwd(dy, mask, 0.5)
print("bwd zero-on-dropped:", (dx[~bits] == 0).all().item())
print("bwd scale maxrel:",
      ((dx[bits] - dy[bits] * 2).abs() /
       (dy[bits].abs() * 2 + 1e-12)).max().item())
print("kept frac:", bits.float().mean().item())

for p in (0.1, 0.5):
    torch.manual_seed(1)
    xx = torch.randn(4000, 130, device="cuda", requires_grad=True)
    yy = ops.fused_dropout(xx, p)
    kept = yy != 0
    g = torch.randn_like(yy)
    yy.backward(g)
    a = xx.grad[kept]
    b = g[kept] / (1 - p)
    rel = (a - b).abs() / (b.abs() + 1e-12)
    print(p, "maxrel", rel.max().item(),
          "dropped-nonzero", (xx.grad[~kept] != 0).sum().item(),
          "frac", kept.float().mean().item())
