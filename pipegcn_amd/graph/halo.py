"""Runtime halo-graph construction for one partition.

Reimplements the per-process graph surgery of the reference
(/root/reference/train.py:84-155, 206-229): train-first renumbering of inner
nodes, halo ("U") node space [inner | halo grouped by owner asc, sorted by
global id], boundary-set ring exchange, recv shapes, and the CSR/CSC pair of
the bipartite halo graph.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from pipegcn_amd.graph.csr import HaloGraph
from pipegcn_amd.graph.partition import PartData
from pipegcn_amd.parallel.transport import exchange_index_lists


@dataclass
class RuntimePartition:
    graph: HaloGraph
    boundary: List[Optional[torch.Tensor]]  # per peer: my inner ids (int64)
    recv_shape: List[Optional[int]]  # per peer: my halo count owned by peer
    ndata: Dict[str, torch.Tensor]  # renumbered inner-node data
    num_in: int
    num_all: int
    num_train: int  # local train-node count (they occupy ids [0, num_train))
    n_feat: int
    n_class: int
    n_train: int  # global train count

    def to(self, device) -> "RuntimePartition":
        self.graph = self.graph.to(device)
        self.boundary = [b.to(device) if b is not None else None
                         for b in self.boundary]
        self.ndata = {k: t.to(device) for k, t in self.ndata.items()}
        return self


def build_runtime_partition(part: PartData, device: str = "cpu",
                            solo: bool = False) -> RuntimePartition:
    """Construct the training-time halo graph for this rank's partition.

    Requires torch.distributed to be initialized (world_size may be 1).

    solo=True: single-process sizing mode — the partition is laid out as if
    its full world (len(node_offsets)-1 ranks) were running, with the real
    halo rows and recv shapes, but no peers exist: the boundary exchange is
    skipped and boundary sets are empty (nothing will be sent). Used by
    `bench.py --solo-of N` for the papers100M 288 GB/GPU sizing run.
    """
    rank = dist.get_rank() if dist.is_initialized() else 0
    size = dist.get_world_size() if dist.is_initialized() else 1
    if solo:
        assert size == 1 and rank == 0, "solo mode is single-process"
        size = len(part.node_offsets) - 1
    num_in = part.num_in
    num_halo = part.halo_gnid.numel()
    num_all = num_in + num_halo
    offsets = part.node_offsets

    # --- train-first renumbering of inner nodes
    # (reference move_train_first, /root/reference/train.py:134-155)
    train_mask = part.ndata["train_mask"]
    num_train = int(train_mask.sum().item())
    new_id = torch.zeros(num_in, dtype=torch.long)
    new_id[train_mask] = torch.arange(num_train)
    new_id[~train_mask] = torch.arange(num_train, num_in)

    ndata = {}
    for k, t in part.ndata.items():
        nt = torch.empty_like(t)
        nt[new_id] = t
        ndata[k] = nt
    # reshuffled global id of each (renumbered) inner node — debugging/tests
    gid = torch.arange(offsets[rank], offsets[rank] + num_in)
    ndata["gid"] = torch.empty_like(gid)
    ndata["gid"][new_id] = gid

    u, v = part.edges[0], part.edges[1]
    u = torch.where(u < num_in, new_id[u.clamp(max=num_in - 1)], u)
    v = new_id[v]

    # --- per-owner halo layout (slots already sorted by reshuffled gid,
    # hence grouped by owner rank ascending)
    recv_shape: List[Optional[int]] = [None] * size
    wanted: List[Optional[torch.Tensor]] = [None] * size
    if num_halo > 0:
        own = torch.bucketize(part.halo_gnid,
                              torch.tensor(offsets[1:], dtype=torch.long),
                              right=True)
    else:
        own = torch.zeros(0, dtype=torch.long)
    for j in range(size):
        if j == rank:
            continue
        sel = own == j
        cnt = int(sel.sum().item())
        recv_shape[j] = cnt
        wanted[j] = part.halo_gnid[sel] - offsets[j]

    # --- boundary exchange (who needs my rows)
    if solo:
        boundary = [torch.zeros(0, dtype=torch.long) for _ in range(size)]
        boundary[rank] = None
    elif size > 1:
        got = exchange_index_lists(wanted)
        boundary: List[Optional[torch.Tensor]] = [
            new_id[g] if g is not None else None for g in got
        ]
        boundary[rank] = None
    else:
        boundary = [None]

    graph = HaloGraph.from_edges(u, v, num_in, num_all)

    rp = RuntimePartition(graph=graph, boundary=boundary,
                          recv_shape=recv_shape, ndata=ndata, num_in=num_in,
                          num_all=num_all, num_train=num_train,
                          n_feat=part.n_feat, n_class=part.n_class,
                          n_train=part.n_train)
    if device != "cpu":
        rp.to(device)
    return rp
