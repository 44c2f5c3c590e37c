"""Graph partitioning + on-disk partition format.

Replaces `dgl.distributed.partition_graph` / `load_partition`
(/root/reference/helper/utils.py:132-144, 99-129) with our own partitioner
(pipegcn_amd._C.partition_graph — BFS-grown + FM-refined, objective
'cut'/'vol'; METIS itself is not available in this environment) and our own
documented partition format.

Format (directory partitions/<graph_name>/):
  meta.json             {num_parts, node_offsets, n_feat, n_class, n_train,
                         multilabel, num_nodes, num_edges}
  part<rank>.pt         torch.save dict with
     num_in       int — number of inner (owned) nodes
     edges        int64 [2, E_local] (u -> v) in LOCAL ids:
                  inner nodes are [0, num_in) ordered by reshuffled global id;
                  halo nodes are [num_in, num_local), sorted by reshuffled
                  global id (== grouped by owner rank asc, sorted within)
     halo_gnid    int64 [num_local - num_in] reshuffled global ids of halo
     feat/label/train_mask/val_mask/test_mask/in_degree — inner nodes only,
                  in local order. in_degree is the FULL-graph in-degree
                  (with self-loops), precomputed before partitioning —
                  reference semantics (/root/reference/helper/utils.py:142).

Reshuffle: like the reference (DGL reshuffle=True), global node ids are
renumbered so partition p owns the contiguous range
[node_offsets[p], node_offsets[p+1]).
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import Dict, List

import torch

from pipegcn_amd import native


@dataclass
class PartData:
    """One rank's partition, as loaded from disk (or built synthetically)."""

    num_in: int
    edges: torch.Tensor  # int64 [2, E] local ids
    halo_gnid: torch.Tensor  # int64 [num_halo] reshuffled global ids
    node_offsets: List[int]  # per-partition global id ranges
    ndata: Dict[str, torch.Tensor]  # feat/label/masks/in_degree (inner only)
    n_feat: int
    n_class: int
    n_train: int  # GLOBAL number of train nodes

    @property
    def num_local(self) -> int:
        return self.num_in + self.halo_gnid.numel()


def assign_partitions(u: torch.Tensor, v: torch.Tensor, num_nodes: int,
                      nparts: int, method: str = "metis",
                      objective: str = "vol", seed: int = 0) -> torch.Tensor:
    """Partition assignment for every node (int32 [N])."""
    if nparts == 1:
        return torch.zeros(num_nodes, dtype=torch.int32)
    if method == "random":
        g = torch.Generator().manual_seed(seed)
        return torch.randint(0, nparts, (num_nodes,), dtype=torch.int32,
                             generator=g)
    if method != "metis":
        raise ValueError(f"unknown partition method: {method}")
    # symmetrize for the partitioner
    su = torch.cat([u, v])
    sv = torch.cat([v, u])
    indptr, indices = native().build_csr(su.cpu(), sv.cpu(), num_nodes)
    obj = 1 if objective == "vol" else 0
    return native().partition_graph(indptr, indices, nparts, obj, 0.05, 8,
                                    seed)


def partition_and_save(u: torch.Tensor, v: torch.Tensor, num_nodes: int,
                       ndata: Dict[str, torch.Tensor], graph_dir: str,
                       nparts: int, method: str = "metis",
                       objective: str = "vol", seed: int = 0) -> None:
    """Partition the global graph and write per-rank files.

    Idempotent like the reference (/root/reference/helper/utils.py:137): skips
    if meta.json already exists.
    """
    meta_path = os.path.join(graph_dir, "meta.json")
    if os.path.exists(meta_path):
        return
    os.makedirs(graph_dir, exist_ok=True)
    u = u.to(torch.long)
    v = v.to(torch.long)

    # full-graph in-degree (with self-loops) BEFORE partitioning
    in_degree = torch.bincount(v, minlength=num_nodes).to(torch.float32)

    part = assign_partitions(u, v, num_nodes, nparts, method, objective,
                             seed).to(torch.long)

    # reshuffle: stable sort nodes by partition -> contiguous global ranges
    order = torch.argsort(part, stable=True)  # old ids in new order
    new_gid = torch.empty(num_nodes, dtype=torch.long)
    new_gid[order] = torch.arange(num_nodes)
    counts = torch.bincount(part, minlength=nparts)
    node_offsets = [0] + torch.cumsum(counts, 0).tolist()

    nu = new_gid[u]
    nv = new_gid[v]
    npart_v = part[v]

    label = ndata["label"]
    multilabel = label.dim() > 1
    n_feat = ndata["feat"].shape[1]
    n_class = label.shape[1] if multilabel else int(label.max().item()) + 1
    n_train = int(ndata["train_mask"].sum().item())

    for p in range(nparts):
        off, end = node_offsets[p], node_offsets[p + 1]
        num_in = end - off
        sel = npart_v == p
        pu, pv = nu[sel], nv[sel]
        local_v = pv - off
        inner_src = (pu >= off) & (pu < end)
        halo_gnid = torch.unique(pu[~inner_src])  # sorted ascending
        # local id map for srcs
        local_u = torch.empty_like(pu)
        local_u[inner_src] = pu[inner_src] - off
        if halo_gnid.numel() > 0:
            halo_pos = torch.searchsorted(halo_gnid, pu[~inner_src])
            local_u[~inner_src] = num_in + halo_pos
        inner_old = order[off:end]  # old ids of this part's inner nodes
        pdata = {
            "num_in": num_in,
            "edges": torch.stack([local_u, local_v]),
            "halo_gnid": halo_gnid,
            "in_degree": in_degree[inner_old],
        }
        for key in ("feat", "label", "train_mask", "val_mask", "test_mask"):
            if key in ndata:
                pdata[key] = ndata[key][inner_old]
        torch.save(pdata, os.path.join(graph_dir, f"part{p}.pt"))

    with open(meta_path, "w") as f:
        json.dump(
            {
                "num_parts": nparts,
                "node_offsets": node_offsets,
                "n_feat": n_feat,
                "n_class": n_class,
                "n_train": n_train,
                "multilabel": multilabel,
                "num_nodes": num_nodes,
                "num_edges": int(u.numel()),
            }, f)


def load_partition(graph_dir: str, rank: int) -> PartData:
    with open(os.path.join(graph_dir, "meta.json")) as f:
        meta = json.load(f)
    d = torch.load(os.path.join(graph_dir, f"part{rank}.pt"),
                   weights_only=True)
    ndata = {k: d[k] for k in
             ("feat", "label", "train_mask", "val_mask", "test_mask",
              "in_degree") if k in d}
    return PartData(
        num_in=d["num_in"],
        edges=d["edges"],
        halo_gnid=d["halo_gnid"],
        node_offsets=meta["node_offsets"],
        ndata=ndata,
        n_feat=meta["n_feat"],
        n_class=meta["n_class"],
        n_train=meta["n_train"],
    )
