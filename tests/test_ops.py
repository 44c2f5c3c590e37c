"""Numerics tests for the HIP kernels vs plain PyTorch fp32 references.

CPU variants cover the native C++ paths; @pytest.mark.gpu variants run the
hand-written gfx950 kernels on a real MI355X.
"""
import pytest
import torch

from pipegcn_amd import ops
from pipegcn_amd.graph.csr import CSR, HaloGraph


def make_graph(n_src=300, n_dst=200, e=4000, f=256, seed=0, device="cpu"):
    g = torch.Generator().manual_seed(seed)
    u = torch.randint(0, n_src, (e,), generator=g)
    v = torch.randint(0, n_dst, (e,), generator=g)
    feat = torch.randn(n_src, f, generator=g)
    hg = HaloGraph.from_edges(u, v, n_dst, n_src).to(device)
    return u, v, hg, feat.to(device)


def torch_spmm(u, v, feat, num_rows, scale=None):
    out = torch.zeros(num_rows, feat.shape[1], device=feat.device)
    out.index_add_(0, v.to(feat.device), feat[u.to(feat.device)])
    if scale is not None:
        out *= scale.unsqueeze(1)
    return out


@pytest.mark.parametrize("f", [4, 41, 100, 256, 602])
def test_spmm_autograd_cpu(f):
    _spmm_autograd_check("cpu", f)


@pytest.mark.gpu
@pytest.mark.parametrize("f", [4, 41, 100, 256, 602])
def test_spmm_autograd_gpu(f):
    _spmm_autograd_check("cuda", f)


def _spmm_autograd_check(device, f):
    u, v, hg, feat = make_graph(f=f, device=device)
    deg = torch.bincount(v, minlength=hg.num_in).float().clamp(min=1)
    inv_deg = (1.0 / deg).to(device)
    feat = feat.requires_grad_(True)
    out = ops.spmm_mean(hg, feat, inv_deg)
    # torch fp32 reference with autograd
    feat_ref = feat.detach().clone().requires_grad_(True)
    ref = torch_spmm(u, v, feat_ref, hg.num_in, inv_deg.cpu().to(device))
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
    gout = torch.randn_like(out)
    out.backward(gout)
    ref.backward(gout)
    assert torch.allclose(feat.grad, feat_ref.grad, atol=1e-4, rtol=1e-4)


@pytest.mark.gpu
def test_spmm_empty_rows_gpu():
    # rows with no in-edges must come out zero, not garbage
    u = torch.tensor([0, 1])
    v = torch.tensor([0, 0])
    hg = HaloGraph.from_edges(u, v, 5, 5).to("cuda")
    feat = torch.randn(5, 7, device="cuda")
    out = ops.spmm(hg.csr, feat, None)
    assert torch.allclose(out[1:], torch.zeros(4, 7, device="cuda"))
    assert torch.allclose(out[0], feat[0] + feat[1])


@pytest.mark.parametrize("device", ["cpu"])
@pytest.mark.parametrize("f", [3, 64, 130])
def test_gather_scatter_cpu(device, f):
    _gather_scatter_check(device, f)


@pytest.mark.gpu
@pytest.mark.parametrize("f", [3, 64, 130])
def test_gather_scatter_gpu(f):
    _gather_scatter_check("cuda", f)


def _gather_scatter_check(device, f):
    src = torch.randn(100, f, device=device)
    idx = torch.randperm(100, device=device)[:30]
    g = ops.gather_rows(src, idx)
    assert torch.allclose(g, src[idx])
    out = torch.empty(30, f, device=device)
    ops.gather_rows_into(src, idx, out)
    assert torch.allclose(out, src[idx])
    dst = torch.randn(100, f, device=device)
    ref = dst.clone()
    add = torch.randn(30, f, device=device)
    ops.scatter_add_rows(dst, idx, add)
    ref[idx] += add
    assert torch.allclose(dst, ref, atol=1e-5)


def test_ema_cpu():
    _ema_check("cpu")


@pytest.mark.gpu
def test_ema_gpu():
    _ema_check("cuda")


def _ema_check(device):
    for numel in (12, 1024, 1027):
        avg = torch.randn(numel, device=device).view(1, -1).contiguous()
        x = torch.randn(numel, device=device).view(1, -1).contiguous()
        ref = 0.95 * avg + 0.05 * x
        ops.ema_update(avg, x, 0.95)
        assert torch.allclose(avg, ref, atol=1e-6)


@pytest.mark.gpu
def test_native_kernels_loaded_on_gpu():
    """The GPU path must run OUR extension, not an eager fallback."""
    import pipegcn_amd

    assert pipegcn_amd.HAS_NATIVE
    assert pipegcn_amd._C.with_hip
    # the .so must live in-tree so it ships with the repo snapshot
    assert "pipegcn_amd" in pipegcn_amd._C.__file__


def test_spmm_src_scale_cpu():
    _src_scale_check("cpu")


@pytest.mark.gpu
def test_spmm_src_scale_gpu():
    _src_scale_check("cuda")


def _src_scale_check(device):
    u, v, hg, feat = make_graph(f=64, device=device)
    ss = (torch.rand(300) + 0.5).to(device)
    out = ops.spmm(hg.csr, feat, None, src_scale=ss)
    ref = torch_spmm(u, v, feat * ss.unsqueeze(1), hg.num_in)
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize("m,k,n", [(1000, 602, 256), (777, 256, 256),
                                   (555, 256, 41), (130, 33, 17)])
def test_sage_dual_gemm_gpu(m, k, n):
    torch.manual_seed(m + k + n)
    x1 = torch.randn(m, k, device="cuda", requires_grad=True)
    x2 = torch.randn(m, k, device="cuda", requires_grad=True)
    l1 = torch.nn.Linear(k, n).cuda()
    l2 = torch.nn.Linear(k, n).cuda()
    out = ops.sage_dual_linear(x1, x2, l1, l2)
    ref = l1(x1) + l2(x2)
    scale = ref.abs().max()
    assert (out - ref).abs().max() / scale < 1e-5, \
        (out - ref).abs().max().item()
    g = torch.randn_like(out)
    out.backward(g, retain_graph=False)
    gx1, gx2 = x1.grad.clone(), x2.grad.clone()
    gw1 = l1.weight.grad.clone()
    x1.grad = x2.grad = None
    l1.weight.grad = l1.bias.grad = None
    ref = l1(x1) + l2(x2)
    ref.backward(g)
    assert torch.allclose(gx1, x1.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(gx2, x2.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(gw1, l1.weight.grad, atol=1e-2, rtol=1e-3)


@pytest.mark.gpu
def test_spmm_bf16_gpu():
    """bf16 SpMM (fp32 accumulate) vs the fp32 kernel."""
    u, v, hg, feat = make_graph(f=256, e=20000, device="cuda")
    deg = hg.csr.row_degrees().to("cuda").clamp(min=1)
    inv = (1.0 / deg).contiguous()
    ref = ops.spmm(hg.csr, feat, inv)
    out = ops.spmm(hg.csr, feat.to(torch.bfloat16), inv)
    assert out.dtype == torch.bfloat16
    err = (out.float() - ref).abs().max() / ref.abs().max()
    assert err < 0.02, err.item()


@pytest.mark.gpu
def test_gather_scatter_bf16_gpu():
    src = torch.randn(100, 130, device="cuda").to(torch.bfloat16)
    idx = torch.randperm(100, device="cuda")[:30]
    g = ops.gather_rows(src, idx)
    assert torch.equal(g, src[idx])
    dst = torch.randn(100, 130, device="cuda").to(torch.bfloat16)
    ref = dst.float().clone()
    add = torch.randn(30, 130, device="cuda").to(torch.bfloat16)
    ops.scatter_add_rows(dst, idx, add)
    ref[idx] += add.float()
    assert torch.allclose(dst.float(), ref, atol=0.05, rtol=0.02)


def test_ema_bf16_fallback():
    avg = torch.randn(64).to(torch.bfloat16)
    x = torch.randn(64).to(torch.bfloat16)
    ref = (0.9 * avg.float() + 0.1 * x.float()).to(torch.bfloat16)
    ops.ema_update(avg, x, 0.9)
    assert torch.allclose(avg.float(), ref.float(), atol=0.05)


@pytest.mark.gpu
@pytest.mark.parametrize("n", [47, 100, 256])
def test_linear_colsum_gpu(n):
    """ops.linear (colsum bias grad) vs torch nn.Linear fwd+bwd."""
    torch.manual_seed(n)
    x = torch.randn(5000, 64, device="cuda", requires_grad=True)
    lin = torch.nn.Linear(64, n).cuda()
    out = ops.linear(x, lin)
    ref = lin(x)
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
    g = torch.randn_like(out)
    out.backward(g)
    gx, gw, gb = x.grad.clone(), lin.weight.grad.clone(), \
        lin.bias.grad.clone()
    x.grad = None
    lin.weight.grad = lin.bias.grad = None
    lin(x).backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(gw, lin.weight.grad, atol=1e-2, rtol=1e-3)
    assert torch.allclose(gb, lin.bias.grad, atol=1e-2, rtol=1e-3)


@pytest.mark.gpu
@pytest.mark.parametrize("f", [128, 256, 602, 100])
@pytest.mark.parametrize("relu", [True, False])
def test_layer_norm_relu_gpu(f, relu):
    """Fused LayerNorm[+ReLU] fwd+bwd vs eager fp32 torch."""
    torch.manual_seed(f)
    n = 3111
    x = torch.randn(n, f, device="cuda", requires_grad=True)
    ln = torch.nn.LayerNorm(f).cuda()
    with torch.no_grad():
        ln.weight.uniform_(0.5, 1.5)
        ln.bias.uniform_(-0.5, 0.5)
    out = ops.layer_norm_relu(x, ln, relu=relu)
    ref = ln(x)
    if relu:
        ref = torch.nn.functional.relu(ref)
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
    g = torch.randn_like(out)
    out.backward(g)
    gx, gw, gb = (x.grad.clone(), ln.weight.grad.clone(),
                  ln.bias.grad.clone())
    x.grad = None
    ln.weight.grad = ln.bias.grad = None
    r = ln(x)
    if relu:
        r = torch.nn.functional.relu(r)
    r.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-4, rtol=1e-3)
    assert torch.allclose(gw, ln.weight.grad, atol=1e-2, rtol=1e-3)
    assert torch.allclose(gb, ln.bias.grad, atol=1e-2, rtol=1e-3)


@pytest.mark.gpu
def test_layer_norm_relu_bf16_gpu():
    """bf16 fused LN+ReLU tracks the fp32 eager reference."""
    torch.manual_seed(0)
    x32 = torch.randn(2048, 256, device="cuda")
    x = x32.to(torch.bfloat16).requires_grad_(True)
    ln = torch.nn.LayerNorm(256).cuda().to(torch.bfloat16)
    out = ops.layer_norm_relu(x, ln, relu=True)
    ref = torch.nn.functional.relu(
        torch.nn.functional.layer_norm(
            x.float(), (256,), ln.weight.float(), ln.bias.float(), ln.eps))
    assert torch.allclose(out.float(), ref, atol=0.05, rtol=0.05)
    out.sum().backward()
    assert torch.isfinite(x.grad.float()).all()
    assert torch.isfinite(ln.weight.grad.float()).all()


@pytest.mark.gpu
@pytest.mark.parametrize("p", [0.1, 0.5])
def test_fused_dropout_gpu(p):
    """Bitmask dropout: scale/zero pattern, statistics, bwd consistency."""
    torch.manual_seed(1)
    x = torch.randn(4000, 130, device="cuda", requires_grad=True)
    y = ops.fused_dropout(x, p)
    # `y != 0` misclassifies an input that is exactly 0.0 as dropped, so
    # restrict the kept/dropped partition to nonzero inputs (the GPU
    # Philox randn does emit exact zeros: ~1 per 5e5 draws observed)
    nz = x.detach() != 0
    kept = (y != 0) & nz
    dropped = (y == 0) & nz
    # kept elements are x/(1-p) (up to p's 1/65536 quantization)
    assert torch.allclose(y[kept], x.detach()[kept] / (1 - p), rtol=1e-3)
    frac = kept.float().mean().item()
    assert abs(frac - (1 - p)) < 0.01, frac
    # backward uses the SAME mask
    g = torch.randn_like(y)
    y.backward(g)
    assert torch.allclose(x.grad[kept], g[kept] / (1 - p), rtol=1e-3)
    assert (x.grad[dropped] == 0).all()


@pytest.mark.gpu
def test_fused_dropout_determinism_gpu():
    """Same torch seed => same mask (counter-based RNG, host-drawn seed)."""
    x = torch.randn(999, 67, device="cuda")
    torch.manual_seed(42)
    y1 = ops.fused_dropout(x, 0.5)
    torch.manual_seed(42)
    y2 = ops.fused_dropout(x, 0.5)
    assert torch.equal(y1, y2)


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(10000, 256, 602), (8192, 256, 256),
                                   (7001, 41, 256), (999, 33, 65)])
def test_dual_wgrad_gpu(shape):
    """Fused MFMA split-M wgrad vs plain fp32 matmul reference."""
    from pipegcn_amd import native

    M, N, K = shape
    torch.manual_seed(0)
    g = torch.randn(M, N, device="cuda")
    x1 = torch.randn(M, K, device="cuda")
    x2 = torch.randn(M, K, device="cuda")
    gw1, gw2 = native().dual_wgrad(g, x1, x2)
    ref1 = g.t() @ x1
    ref2 = g.t() @ x2
    assert torch.allclose(gw1, ref1, rtol=1e-4, atol=1e-2), \
        (gw1 - ref1).abs().max()
    assert torch.allclose(gw2, ref2, rtol=1e-4, atol=1e-2)
    # single-input variant (tail linears, GCN)
    (gws,) = native().dual_wgrad(g, x1, torch.Tensor())
    assert torch.allclose(gws, ref1, rtol=1e-4, atol=1e-2)
    # determinism: the split-M reduce must be bitwise reproducible
    gw1b, _ = native().dual_wgrad(g, x1, x2)
    assert torch.equal(gw1, gw1b)


@pytest.mark.gpu
def test_dual_linear_backward_uses_fused_wgrad():
    """End-to-end: _SageDualLinear backward grads match eager autograd."""
    torch.manual_seed(1)
    M, K, N = 5000, 130, 96
    lin1 = torch.nn.Linear(K, N).cuda()
    lin2 = torch.nn.Linear(K, N).cuda()
    x1 = torch.randn(M, K, device="cuda", requires_grad=True)
    x2 = torch.randn(M, K, device="cuda", requires_grad=True)
    out = ops.sage_dual_linear(x1, x2, lin1, lin2)
    gout = torch.randn_like(out)
    out.backward(gout)
    # eager reference
    x1r = x1.detach().clone().requires_grad_(True)
    x2r = x2.detach().clone().requires_grad_(True)
    l1r = torch.nn.Linear(K, N).cuda()
    l2r = torch.nn.Linear(K, N).cuda()
    l1r.load_state_dict(lin1.state_dict())
    l2r.load_state_dict(lin2.state_dict())
    (l1r(x1r) + l2r(x2r)).backward(gout)
    for a, b in ((x1.grad, x1r.grad), (x2.grad, x2r.grad),
                 (lin1.weight.grad, l1r.weight.grad),
                 (lin2.weight.grad, l2r.weight.grad),
                 (lin1.bias.grad, l1r.bias.grad)):
        assert torch.allclose(a, b, rtol=1e-4, atol=1e-2), \
            (a - b).abs().max()


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(10000, 256, 602), (8192, 256, 256),
                                   (999, 64, 100), (5000, 100, 70)])
def test_dual_dgrad_gpu(shape):
    """Fused MFMA dual-dgrad vs plain fp32 matmul reference."""
    from pipegcn_amd import native

    M, N, K = shape
    torch.manual_seed(0)
    g = torch.randn(M, N, device="cuda")
    w1 = torch.randn(N, K, device="cuda")
    w2 = torch.randn(N, K, device="cuda")
    gx1, gx2 = native().dual_dgrad(g, w1, w2)
    assert torch.allclose(gx1, g @ w1, rtol=1e-4, atol=1e-2), \
        (gx1 - g @ w1).abs().max()
    assert torch.allclose(gx2, g @ w2, rtol=1e-4, atol=1e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("f", [33, 602, 1000, 1024])
def test_layer_norm_relu_odd_widths_gpu(f):
    """LN kernel edge sizes: odd F, F not a multiple of 64*VEC, the
    F<=1024 register cap boundary."""
    torch.manual_seed(f)
    x = torch.randn(777, f, device="cuda", requires_grad=True)
    ln = torch.nn.LayerNorm(f).cuda()
    out = ops.layer_norm_relu(x, ln, relu=True)
    ref = torch.nn.functional.relu(
        torch.nn.functional.layer_norm(x, (f,), ln.weight, ln.bias, ln.eps))
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
    g = torch.randn_like(out)
    out.backward(g)
    gx = x.grad.clone()
    x.grad = None
    ln.weight.grad = ln.bias.grad = None
    torch.nn.functional.relu(
        torch.nn.functional.layer_norm(x, (f,), ln.weight, ln.bias,
                                       ln.eps)).backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-3, rtol=1e-3)


@pytest.mark.gpu
@pytest.mark.parametrize("p", [0.01, 0.9])
def test_fused_dropout_extreme_p_gpu(p):
    torch.manual_seed(0)
    x = torch.randn(100001, device="cuda")  # odd length: tail path
    from pipegcn_amd import native
    y, mask = native().dropout_fwd(x, p, 7)
    frac = (y != 0).float().mean().item()
    assert abs(frac - (1 - p)) < 0.02, frac
    dx = native().dropout_bwd(y, mask, p)
    assert dx.shape == x.shape
