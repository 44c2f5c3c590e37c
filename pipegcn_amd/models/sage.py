"""GraphSAGE model + layer on the native SpMM.

Reimplements /root/reference/module/layer.py and module/model.py on our
HaloGraph/FullGraph CSR structures and hand-written gfx950 SpMM:

 - training path: mean aggregation over the bipartite halo graph with the
   degree-divide fused into the SpMM epilogue, then
   linear1(feat[:num_dst]) + linear2(ah)   (layer.py:44-51);
 - use_pp first layer collapses to one GEMM on [feat ‖ mean_agg]
   (layer.py:41-42);
 - eval path: full homogeneous graph, degrees from the graph (layer.py:52-62);
 - init: uniform ±1/sqrt(fan_in) (layer.py:24-36).
"""
from __future__ import annotations

import math

import torch
from torch import nn

from pipegcn_amd import ops
from pipegcn_amd.graph.csr import FullGraph, HaloGraph
from pipegcn_amd.models.sync_bn import SyncBatchNorm
from pipegcn_amd.parallel import context as ctx


class GraphSAGELayer(nn.Module):
    def __init__(self, in_feats, out_feats, bias=True, use_pp=False):
        super().__init__()
        self.use_pp = use_pp
        if use_pp:
            self.linear = nn.Linear(2 * in_feats, out_feats, bias=bias)
        else:
            self.linear1 = nn.Linear(in_feats, out_feats, bias=bias)
            self.linear2 = nn.Linear(in_feats, out_feats, bias=bias)
        self.reset_parameters()

    def reset_parameters(self):
        def init(lin):
            stdv = 1.0 / math.sqrt(lin.weight.size(1))
            lin.weight.data.uniform_(-stdv, stdv)
            if lin.bias is not None:
                lin.bias.data.uniform_(-stdv, stdv)

        if self.use_pp:
            init(self.linear)
        else:
            init(self.linear1)
            init(self.linear2)

    def forward(self, graph, feat, in_deg=None):
        if self.training:
            if self.use_pp:
                return self.linear(feat)
            assert isinstance(graph, HaloGraph)
            inv_deg = (1.0 / in_deg.clamp(min=1.0)).contiguous()
            ah = ops.spmm_mean(graph, feat, inv_deg)
            x1 = feat[: graph.num_in]
            if feat.is_cuda:
                # hand-written MFMA fused dual GEMM (one output pass)
                return ops.sage_dual_linear(x1, ah, self.linear1,
                                            self.linear2)
            return self.linear1(x1) + self.linear2(ah)
        # eval: full homogeneous graph, degrees from the graph itself
        assert isinstance(graph, FullGraph) and in_deg is None
        degs = graph.in_degrees().clamp(min=1.0)
        ah = ops.spmm(graph.csr, feat, (1.0 / degs).contiguous())
        if self.use_pp:
            return self.linear(torch.cat((feat, ah), dim=1))
        return self.linear1(feat) + self.linear2(ah)


class GNNBase(nn.Module):
    def __init__(self, layer_size, activation, use_pp=False, dropout=0.5,
                 norm="layer", n_linear=0):
        super().__init__()
        self.n_layers = len(layer_size) - 1
        self.layers = nn.ModuleList()
        self.activation = activation
        self.use_pp = use_pp
        self.n_linear = n_linear
        self.use_norm = norm is not None
        if self.use_norm:
            self.norm = nn.ModuleList()
        self.dropout = nn.Dropout(p=dropout)

    def _drop(self, h):
        """Dropout: fused bitmask kernel on GPU, eager elsewhere."""
        if self.training and h.is_cuda and self.dropout.p > 0:
            return ops.fused_dropout(h, self.dropout.p)
        return self.dropout(h)

    def _norm_act(self, i, h):
        """Inter-layer norm + activation: fused LayerNorm+ReLU on GPU."""
        if h.is_cuda and self.use_norm and isinstance(self.norm[i],
                                                      nn.LayerNorm):
            fuse_relu = self.activation is torch.nn.functional.relu
            h = ops.layer_norm_relu(h, self.norm[i], relu=fuse_relu)
            return h if fuse_relu else self.activation(h)
        if self.use_norm:
            h = self.norm[i](h)
        return self.activation(h)


class GraphSAGE(GNNBase):
    def __init__(self, layer_size, activation, use_pp, dropout=0.5,
                 norm="layer", train_size=None, n_linear=0):
        super().__init__(layer_size, activation, use_pp, dropout, norm,
                         n_linear)
        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                self.layers.append(
                    GraphSAGELayer(layer_size[i], layer_size[i + 1],
                                   use_pp=use_pp))
            else:
                self.layers.append(nn.Linear(layer_size[i],
                                             layer_size[i + 1]))
            if i < self.n_layers - 1 and self.use_norm:
                if norm == "layer":
                    self.norm.append(
                        nn.LayerNorm(layer_size[i + 1],
                                     elementwise_affine=True))
                elif norm == "batch":
                    self.norm.append(SyncBatchNorm(layer_size[i + 1],
                                                   train_size))
            use_pp = False

    def forward(self, g, feat, in_deg=None):
        h = feat
        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                if self.training and (i > 0 or not self.use_pp):
                    h = ctx.buffer.update(i, h)
                h = self._drop(h)
                h = self.layers[i](g, h, in_deg)
            else:
                h = self._drop(h)
                h = ops.linear(h, self.layers[i])
            if i < self.n_layers - 1:
                h = self._norm_act(i, h)
        return h
