"""Autograd wrappers over the native gfx950 kernels.

Replaces DGL's `update_all(fn.copy_src, fn.sum)` gSpMM + autograd
(/root/reference/module/layer.py:47-49) with our own CSR SpMM whose backward
runs the pre-built transpose (CSC) through the same kernel — no atomics.

Device dispatch happens inside pipegcn_amd._C: CUDA tensors go to the
hand-written HIP kernels (and FAIL if the extension is absent — there is no
eager fallback on GPU); CPU tensors use the native C++ paths.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from pipegcn_amd import native
from pipegcn_amd.graph.csr import CSR, HaloGraph


def _tune_row_order(csr: CSR, feat, s, ss) -> bool:
    """One-shot A/B of LPT vs natural row order for this (graph, F).

    The best order depends on BOTH graph structure and feature width
    (measured: LPT wins on uniform-source graphs and at narrow F; natural
    order wins by ~11% at F=602 on locality-structured graphs, where the
    degree sort scatters the near-diagonal source reuse). Results are
    BITWISE IDENTICAL either way — row_order only changes which wave
    processes which row, never the per-row edge order — so the choice is
    purely a throughput decision, amortized over thousands of epochs.
    """
    import time

    nat = torch.Tensor()
    times = {}
    for key, ro in (("lpt", csr.row_order), ("nat", nat)):
        native().spmm(csr.indptr, csr.indices, feat, s, ss, ro,
                      csr.num_rows)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(2):
            native().spmm(csr.indptr, csr.indices, feat, s, ss, ro,
                          csr.num_rows)
        torch.cuda.synchronize()
        times[key] = time.perf_counter() - t0
    return times["lpt"] <= times["nat"]


def spmm(csr: CSR, feat: torch.Tensor,
         scale: Optional[torch.Tensor] = None,
         src_scale: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out[r,:] = scale[r] * sum_{u in N(r)} src_scale[u] * feat[u,:]."""
    if feat.shape[0] < csr.num_cols:
        raise ValueError(
            f"spmm: feat has {feat.shape[0]} rows but the graph references "
            f"{csr.num_cols} source nodes (an under-sized feat would be an "
            "out-of-bounds GPU gather)")
    s = scale if scale is not None else torch.Tensor()
    ss = src_scale if src_scale is not None else torch.Tensor()
    feat = feat.contiguous()
    ro = csr.row_order if csr.row_order is not None else torch.Tensor()
    # auto-tuned row order on real-sized GPU workloads (>= ~1G gathered
    # elements, where a kernel is multi-ms and the 6 extra timed launches
    # vanish into warmup); PIPEGCN_SPMM_AUTOTUNE=0 pins LPT
    if (feat.is_cuda and csr.row_order is not None
            and csr.nnz * feat.shape[1] >= (1 << 30)
            and os.environ.get("PIPEGCN_SPMM_AUTOTUNE", "1") == "1"):
        cache = getattr(csr, "_lpt_choice", None)
        if cache is None:
            cache = {}
            csr._lpt_choice = cache
        use_lpt = cache.get(feat.shape[1])
        if use_lpt is None:
            # the A/B makes 6 transient outputs; near the memory ceiling
            # (papers100M sizing: tuning fires mid-backward at peak) that
            # transient OOMs or inflates peak — keep the LPT default there
            free, _ = torch.cuda.mem_get_info(feat.device)
            out_bytes = csr.num_rows * feat.shape[1] * feat.element_size()
            if free < 3 * out_bytes + (2 << 30):
                use_lpt = True
            else:
                use_lpt = _tune_row_order(csr, feat, s, ss)
            cache[feat.shape[1]] = use_lpt
        if not use_lpt:
            ro = torch.Tensor()
    return native().spmm(csr.indptr, csr.indices, feat, s, ss,
                         ro, csr.num_rows)


class _SpmmMean(torch.autograd.Function):
    """ah = D^{-1} A h over the halo graph; backward = A^T D^{-1} g."""

    @staticmethod
    def forward(ctx, graph: HaloGraph, feat: torch.Tensor,
                inv_deg: torch.Tensor):
        ctx.graph = graph
        ctx.save_for_backward(inv_deg)
        return spmm(graph.csr, feat, inv_deg)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (inv_deg,) = ctx.saved_tensors
        g = ctx.graph
        # D^{-1} either as ONE streaming row-multiply on g or as the
        # fused per-EDGE src_scale gather: the multiply costs ~3 passes
        # over [rows,F], the gather ~one 4B load per edge plus its
        # latency slot — dense graphs (reddit: nnz 115M > numel 60M,
        # measured +1 ms/call for pre-scale) want the multiply, sparse
        # wide ones (yelp: nnz 14M << numel 367M) want the gather
        if g.csc.nnz > grad_out.numel():
            grad_feat = spmm(
                g.csc,
                grad_out * inv_deg.unsqueeze(1).to(grad_out.dtype), None)
        else:
            grad_feat = spmm(g.csc, grad_out.contiguous(), None,
                             src_scale=inv_deg)
        return None, grad_feat, None


def spmm_mean(graph: HaloGraph, feat: torch.Tensor,
              inv_deg: torch.Tensor) -> torch.Tensor:
    """Differentiable mean-aggregation over the halo graph.

    inv_deg is 1/in_degree of the dst nodes (full-graph degrees, precomputed
    before partitioning — reference semantics, /root/reference/helper/utils.py:142).
    """
    return _SpmmMean.apply(graph, feat, inv_deg)


def gather_rows(src: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    return native().gather_rows(src.contiguous(), idx)


def gather_rows_into(src: torch.Tensor, idx: torch.Tensor,
                     out: torch.Tensor) -> None:
    """Gather into a persistent buffer (avoids per-epoch allocation)."""
    if src.is_cuda:
        native().gather_rows_out(src.contiguous(), idx, out)
    else:
        torch.index_select(src, 0, idx, out=out)


def scatter_add_rows(dst: torch.Tensor, idx: torch.Tensor,
                     src: torch.Tensor) -> None:
    """dst[idx[i],:] += src[i,:]; idx must be unique."""
    native().scatter_add_rows(dst, idx, src.contiguous())


def ema_update(avg: torch.Tensor, x: torch.Tensor, momentum: float) -> None:
    """avg = momentum * avg + (1 - momentum) * x (in place)."""
    if avg.dtype != torch.float32:
        avg.mul_(momentum).add_(x, alpha=1.0 - momentum)
        return
    native().ema_update(avg, x, momentum)


class _SageDualLinear(torch.autograd.Function):
    """out = x1 @ w1^T + x2 @ w2^T + (b1 + b2).

    The whole dense path is hand-written gfx950 MFMA: forward fused
    dual-GEMM (csrc/hip/dual_gemm.hip), backward fused dual-dgrad
    (dual_dgrad.hip) + fused split-M dual-wgrad (wgrad.hip) + native
    colsum bias grad; rocBLAS only on thin-N (output-layer) shapes where
    the 32-wide MFMA tiles would idle.
    """

    @staticmethod
    def forward(ctx, x1, x2, w1, w2, b1, b2):
        ctx.save_for_backward(x1, x2, w1, w2)
        ctx.has_bias = b1 is not None
        if (w1.size(0) < 64 or not x1.is_cuda
                or x1.dtype != torch.float32):
            # thin-N / CPU: the 128x128 MFMA tile would waste most of the
            # block; rocBLAS pair instead (the op is memory-bound there)
            out = torch.mm(x1, w1.t())
            out.addmm_(x2, w2.t())
            if b1 is not None:
                out.add_(b1 + b2)
            return out
        bias = (b1 + b2) if b1 is not None else torch.Tensor()
        return native().sage_dual_gemm(x1.contiguous(), x2.contiguous(),
                                       w1.contiguous(), w2.contiguous(),
                                       bias)

    @staticmethod
    def backward(ctx, g):
        x1, x2, w1, w2 = ctx.saved_tensors
        g = g.contiguous()
        if g.is_cuda and g.dtype == torch.float32:
            # NOTE: running the wgrad pair on a side stream under the
            # dgrad was tried and measured WORSE (113.7 -> 114.7 ms
            # epoch): the timeline is gap-free compute, so concurrent
            # saturating kernels just share CUs plus event overhead.
            # wgrad pair fused in one MFMA split-M kernel (g streamed
            # once for both products; deterministic workspace reduce)
            gw1, gw2 = native().dual_wgrad(g, x1.contiguous(),
                                           x2.contiguous())
            if g.size(1) >= 64 and w1.size(1) <= 256:
                # dgrad pair fused in one MFMA kernel (g tile staged once
                # for both weight contractions; measured par with rocBLAS
                # at K=256 — 117 TF both; rocBLAS wins at K=512 (142 vs
                # 123 TF) and K=602 (112 vs 95), hence the cap)
                gx1, gx2 = native().dual_dgrad(g, w1.contiguous(),
                                               w2.contiguous())
            else:
                # thin-N (the 41-class output layer) and wide-K (602:
                # rocBLAS 112 vs our 95 TF — the unaligned 602-float rows
                # and boundary k-tile cost us there): ONE GEMM against
                # [w1 ‖ w2] so g is still read once
                gx = g @ torch.cat((w1, w2), dim=1)
                K = w1.size(1)
                gx1, gx2 = gx[:, :K], gx[:, K:]
        else:
            gx1 = g @ w1
            gx2 = g @ w2
            gw1 = g.t() @ x1
            gw2 = g.t() @ x2
        if not ctx.has_bias:
            return gx1, gx2, gw1, gw2, None, None
        gb = native().colsum(g)  # two-phase column sum (fastest measured)
        gb = gb.to(w1.dtype)
        return gx1, gx2, gw1, gw2, gb, gb


def sage_dual_linear(x1, x2, lin1, lin2):
    """Fused linear1(x1) + linear2(x2) for the SAGE layer.

    All shapes go through _SageDualLinear so the bias grad uses the native
    colsum — torch's nn.Linear backward picks a pathological 1024-thread
    reduce for thin odd N ([2.45M,47] bias grad measured 18.7 ms vs 0.2).
    The forward dispatches thin-N / CPU to rocBLAS internally.
    """
    return _SageDualLinear.apply(x1, x2, lin1.weight, lin2.weight, lin1.bias,
                                 lin2.bias)


class _FusedDropout(torch.autograd.Function):
    """Dropout with a BITPACKED mask (1 bit/element vs torch's byte mask —
    8x less saved memory) and the scale fused into the same pass. The RNG is
    counter-based (splitmix64 keyed on a host-drawn seed), so the op is
    reproducible given torch.manual_seed. GPU path only."""

    @staticmethod
    def forward(ctx, x, p, seed):
        y, mask = native().dropout_fwd(x.contiguous(), p, seed)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        return native().dropout_bwd(dy, mask, ctx.p), None, None


def fused_dropout(x: torch.Tensor, p: float) -> torch.Tensor:
    """Training-mode dropout on GPU (call only when training and p > 0)."""
    # host-drawn 63-bit seed: deterministic under torch.manual_seed
    seed = int(torch.randint(0, 2 ** 62, (1,)).item())
    return _FusedDropout.apply(x, p, seed)


class _LayerNormReLU(torch.autograd.Function):
    """Fused LayerNorm [+ReLU] (replaces the eager nn.LayerNorm + F.relu
    pair between layers, /root/reference/module/model.py:53-56).

    Saves xhat + rstd only — eager autograd keeps {LN input, ReLU output}
    (two [N,F] tensors); this keeps one. Statistics in fp32 for both dtypes;
    the ReLU mask is recomputed in backward from w*xhat+b.
    """

    @staticmethod
    def forward(ctx, x, w, b, eps, relu):
        y, xhat, rstd = native().layer_norm_relu_fwd(
            x.contiguous(), w.contiguous(), b.contiguous(), eps, relu)
        ctx.save_for_backward(xhat, rstd, w, b)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        xhat, rstd, w, b = ctx.saved_tensors
        dx, dw, db = native().layer_norm_relu_bwd(dy, xhat, rstd, w, b,
                                                  ctx.relu)
        return dx, dw, db, None, None


def layer_norm_relu(x: torch.Tensor, ln: torch.nn.LayerNorm,
                    relu: bool) -> torch.Tensor:
    """ln(x) [+ relu] through the fused gfx950 kernel (GPU path)."""
    # fp32 weight views keep the kernel dtype-simple; the casts are autograd
    # nodes, so bf16 params still receive their grads (no-op for fp32)
    return _LayerNormReLU.apply(x, ln.weight.float(), ln.bias.float(),
                                ln.eps, relu)


class _LinearColsum(torch.autograd.Function):
    """nn.Linear forward via rocBLAS with a colsum bias grad — torch's
    Linear backward picks a 1024-thread reduce for thin odd N (18.7 ms at
    [2.45M,47]; profiles/README.md)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        out = torch.mm(x, w.t())
        if b is not None:
            out.add_(b)
        return out

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        g = g.contiguous()
        gb = native().colsum(g).to(w.dtype)
        if g.dtype == torch.float32:
            (gw,) = native().dual_wgrad(g, x.contiguous(), torch.Tensor())
        else:
            gw = g.t() @ x
        return g @ w, gw, gb


def linear(x, lin):
    """lin(x) with the fast bias-grad path on GPU."""
    if not x.is_cuda or lin.bias is None:
        return lin(x)
    return _LinearColsum.apply(x, lin.weight, lin.bias)
