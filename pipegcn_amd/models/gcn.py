"""GCN model (symmetric-normalized aggregation) — a capability extension
over the reference (whose `create_model` supports only graphsage,
/root/reference/train.py:192-197); the north star names the "GraphSAGE/GCN
layer hot path".

Layer: H' = D^{-1/2} (A+I) D^{-1/2} H W + b, computed by the SAME gfx950
SpMM kernel with BOTH fused scales (dst_scale = deg(dst)^{-1/2},
src_scale = deg(src)^{-1/2}); self-loops are already present in the graphs
(datasets.py normalization). The src scale needs the in-degrees of HALO
nodes, fetched once at setup (trainer.exchange_halo_values).
"""
from __future__ import annotations

import math

import torch
from torch import nn

from pipegcn_amd import ops
from pipegcn_amd.graph.csr import FullGraph, HaloGraph
from pipegcn_amd.models.sage import GNNBase
from pipegcn_amd.parallel import context as ctx


class _SpmmSym(torch.autograd.Function):
    """sym-normalized aggregation; backward is the transpose with the
    scales swapped (S = D_dst^{-1/2} A D_src^{-1/2};
    S^T = D_src^{-1/2} A^T D_dst^{-1/2})."""

    @staticmethod
    def forward(ctx_, graph: HaloGraph, feat, inv_sqrt_dst, inv_sqrt_all):
        ctx_.graph = graph
        ctx_.save_for_backward(inv_sqrt_dst, inv_sqrt_all)
        # source-side scale: streaming row-multiply on dense graphs,
        # fused per-edge gather on sparse-wide ones (cost model in
        # _SpmmMean.backward)
        if graph.csr.nnz > feat.numel():
            return ops.spmm(graph.csr,
                            feat * inv_sqrt_all.unsqueeze(1).to(feat.dtype),
                            inv_sqrt_dst)
        return ops.spmm(graph.csr, feat, inv_sqrt_dst,
                        src_scale=inv_sqrt_all)

    @staticmethod
    def backward(ctx_, grad_out):
        inv_sqrt_dst, inv_sqrt_all = ctx_.saved_tensors
        g = ctx_.graph
        if g.csc.nnz > grad_out.numel():
            grad_feat = ops.spmm(
                g.csc,
                grad_out * inv_sqrt_dst.unsqueeze(1).to(grad_out.dtype),
                inv_sqrt_all)
        else:
            grad_feat = ops.spmm(g.csc, grad_out.contiguous(),
                                 inv_sqrt_all, src_scale=inv_sqrt_dst)
        return None, grad_feat, None, None


class GCNLayer(nn.Module):
    def __init__(self, in_feats, out_feats, bias=True):
        super().__init__()
        self.linear = nn.Linear(in_feats, out_feats, bias=bias)
        stdv = 1.0 / math.sqrt(self.linear.weight.size(1))
        self.linear.weight.data.uniform_(-stdv, stdv)
        if bias:
            self.linear.bias.data.uniform_(-stdv, stdv)

    def forward(self, graph, feat, deg=None):
        if self.training:
            assert isinstance(graph, HaloGraph)
            inv_sqrt_all = torch.rsqrt(deg.clamp(min=1.0)).contiguous()
            ah = _SpmmSym.apply(graph, feat,
                                inv_sqrt_all[: graph.num_in].contiguous(),
                                inv_sqrt_all)
            return ops.linear(ah, self.linear)
        assert isinstance(graph, FullGraph) and deg is None
        d = torch.rsqrt(graph.in_degrees().clamp(min=1.0)).contiguous()
        ah = ops.spmm(graph.csr, feat, d, src_scale=d)
        return self.linear(ah)


class GCN(GNNBase):
    """Same structure as GraphSAGE: conv layers then linear tail, norm+ReLU
    between layers, `ctx.buffer.update` as the only distributed seam.

    `in_deg` passed to forward must cover ALL local nodes (inner + halo) —
    the trainer assembles it via the one-shot halo degree exchange.
    """

    def __init__(self, layer_size, activation, use_pp=False, dropout=0.5,
                 norm="layer", train_size=None, n_linear=0):
        if use_pp:
            raise NotImplementedError("--use-pp supports graphsage only "
                                      "(reference parity)")
        super().__init__(layer_size, activation, False, dropout, norm,
                         n_linear)
        from pipegcn_amd.models.sync_bn import SyncBatchNorm

        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                self.layers.append(GCNLayer(layer_size[i],
                                            layer_size[i + 1]))
            else:
                self.layers.append(nn.Linear(layer_size[i],
                                             layer_size[i + 1]))
            if i < self.n_layers - 1 and self.use_norm:
                if norm == "layer":
                    self.norm.append(nn.LayerNorm(layer_size[i + 1],
                                                  elementwise_affine=True))
                elif norm == "batch":
                    self.norm.append(SyncBatchNorm(layer_size[i + 1],
                                                   train_size))

    def forward(self, g, feat, in_deg=None):
        h = feat
        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                if self.training:
                    h = ctx.buffer.update(i, h)
                h = self._drop(h)
                h = self.layers[i](g, h, in_deg)
            else:
                h = self._drop(h)
                h = ops.linear(h, self.layers[i])
            if i < self.n_layers - 1:
                h = self._norm_act(i, h)
        return h
