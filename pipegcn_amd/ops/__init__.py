"""Autograd wrappers over the native gfx950 kernels.

Replaces DGL's `update_all(fn.copy_src, fn.sum)` gSpMM + autograd
(/root/reference/module/layer.py:47-49) with our own CSR SpMM whose backward
runs the pre-built transpose (CSC) through the same kernel — no atomics.

Device dispatch happens inside pipegcn_amd._C: CUDA tensors go to the
hand-written HIP kernels (and FAIL if the extension is absent — there is no
eager fallback on GPU); CPU tensors use the native C++ paths.
"""
from __future__ import annotations

from typing import Optional

import torch

from pipegcn_amd import native
from pipegcn_amd.graph.csr import CSR, HaloGraph


def spmm(csr: CSR, feat: torch.Tensor,
         scale: Optional[torch.Tensor] = None,
         src_scale: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out[r,:] = scale[r] * sum_{u in N(r)} src_scale[u] * feat[u,:]."""
    s = scale if scale is not None else torch.Tensor()
    ss = src_scale if src_scale is not None else torch.Tensor()
    return native().spmm(csr.indptr, csr.indices, feat.contiguous(), s, ss,
                         csr.num_rows)


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
        # transpose SpMM with the D^{-1} pre-scale fused as a source scale
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
    native().ema_update(avg, x, momentum)
