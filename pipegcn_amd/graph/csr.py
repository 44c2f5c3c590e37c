"""CSR graph structures.

Replaces the reference's DGL graph objects (`dgl.graph` / the `('_U','_E','_V')`
halo heterograph built at /root/reference/train.py:206-229) with plain CSR
tensors owned by this framework. The halo graph is bipartite:

    src space ("U"): [inner nodes (train-first renumbered) | halo nodes,
                      grouped by owner rank ascending, each group sorted by
                      global id]
    dst space ("V"): the inner nodes (same renumbering)

Both the dst-major CSR (forward SpMM) and its transpose CSC (backward SpMM)
are built once at setup so the backward pass needs no atomics.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from pipegcn_amd import native


@dataclass
class CSR:
    """CSR adjacency over rows; indices are column ids.

    row_order is the LPT schedule (rows sorted by degree descending): the
    SpMM kernel processes heavy lognormal-tail rows first so light rows
    backfill the wavefront slots.
    """

    indptr: torch.Tensor  # int64 [num_rows + 1]
    indices: torch.Tensor  # int32 [nnz]
    num_rows: int
    num_cols: int
    row_order: Optional[torch.Tensor] = None  # int32 [num_rows]

    @staticmethod
    def from_coo(u: torch.Tensor, v: torch.Tensor, num_rows: int,
                 num_cols: int) -> "CSR":
        """Build CSR with rows = v (dst), cols = u (src)."""
        indptr, indices = native().build_csr(u.cpu(), v.cpu(), num_rows)
        deg = indptr[1:] - indptr[:-1]
        order = torch.argsort(deg, descending=True).to(torch.int32)
        return CSR(indptr, indices, num_rows, num_cols, order)

    def to(self, device) -> "CSR":
        return CSR(self.indptr.to(device), self.indices.to(device),
                   self.num_rows, self.num_cols,
                   self.row_order.to(device)
                   if self.row_order is not None else None)

    @property
    def nnz(self) -> int:
        return self.indices.numel()

    def row_degrees(self) -> torch.Tensor:
        return (self.indptr[1:] - self.indptr[:-1]).to(torch.float32)


@dataclass
class HaloGraph:
    """Bipartite halo graph of one partition (training-path structure)."""

    csr: CSR  # rows = inner dst nodes, cols in [0, num_all)
    csc: CSR  # rows = all src nodes (inner + halo), cols = inner dst nodes
    num_in: int
    num_all: int

    def to(self, device) -> "HaloGraph":
        return HaloGraph(self.csr.to(device), self.csc.to(device),
                         self.num_in, self.num_all)

    @staticmethod
    def from_edges(u: torch.Tensor, v: torch.Tensor, num_in: int,
                   num_all: int) -> "HaloGraph":
        """u in [0, num_all), v in [0, num_in) — edges u -> v."""
        csr = CSR.from_coo(u, v, num_in, num_all)
        csc = CSR.from_coo(v, u, num_all, num_in)
        return HaloGraph(csr, csc, num_in, num_all)


@dataclass
class FullGraph:
    """Whole-graph structure for (rank-0, CPU) full-graph evaluation.

    Mirrors the reference's eval path (/root/reference/module/layer.py:52-62):
    homogeneous graph, degrees taken from the graph itself.
    """

    csr: CSR  # rows = dst nodes over the full graph
    num_nodes: int
    ndata: dict  # feat/label/masks

    @staticmethod
    def from_coo(u: torch.Tensor, v: torch.Tensor, num_nodes: int,
                 ndata: Optional[dict] = None) -> "FullGraph":
        return FullGraph(CSR.from_coo(u, v, num_nodes, num_nodes), num_nodes,
                         ndata or {})

    def in_degrees(self) -> torch.Tensor:
        return self.csr.row_degrees()
