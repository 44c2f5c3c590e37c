"""Pipelined (stale) training vs an INDEPENDENT sequential simulator.

The distributed pipelined run (threads + comm stream + ring transport) must
produce, epoch by epoch, exactly the losses of a single-process simulator
that implements the PipeGCN staleness semantics directly (reference
/root/reference/helper/feature_buffer.py):

 - epoch E's forward consumes boundary features sent during epoch E-1
   (zeros at epoch 0);
 - epoch E's backward adds boundary gradients sent during epoch E-1
   (zeros at epoch 0) and records this epoch's halo grads for the peer;
 - gradients are divided by the global train count before summing
   across partitions; Adam steps the replicated model.

This is the executable spec of the protocol — any reordering /
use-after-free / off-by-one-epoch bug in the Buffer shows up here as a
loss divergence.
"""
import os

import torch

from tests.conftest import run_distributed
from tests.test_distributed import WORLD, _prepare_partitions, make_args


def _renumber(part):
    """Replicate halo.py's train-first renumbering (no dist needed)."""
    num_in = part.num_in
    train = part.ndata["train_mask"]
    nt = int(train.sum())
    new_id = torch.zeros(num_in, dtype=torch.long)
    new_id[train] = torch.arange(nt)
    new_id[~train] = torch.arange(nt, num_in)
    ndata = {}
    for k, t in part.ndata.items():
        nd = torch.empty_like(t)
        nd[new_id] = t
        ndata[k] = nd
    u, v = part.edges[0], part.edges[1]
    u = torch.where(u < num_in, new_id[u.clamp(max=num_in - 1)], u)
    v = new_id[v]
    return new_id, ndata, u, v, nt


def _simulate(parts, n_epochs, hidden, lr, n_train_global, seed,
              corr=False, momentum=0.8):
    """Sequential reference implementation of pipelined 2-partition
    training (GraphSAGE, dropout 0, no norm, 2 conv layers); with
    corr=True the EMA smoothing correction replaces the raw stale values
    on both directions (reference feature_buffer.py:186-191,137-140)."""
    import torch.nn.functional as F

    from pipegcn_amd.graph.csr import HaloGraph
    from pipegcn_amd.models.sage import GraphSAGE
    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.parallel.buffer import Buffer

    ctx.buffer = Buffer()  # world-1 (uninitialized paths unused here)

    world = len(parts)
    offs = parts[0].node_offsets
    state = []
    for r, part in enumerate(parts):
        new_id, ndata, u, v, nt = _renumber(part)
        g = HaloGraph.from_edges(u, v, part.num_in, part.num_local)
        # halo slots are sorted by gid => grouped by owner ascending
        own = torch.bucketize(part.halo_gnid,
                              torch.tensor(offs[1:]), right=True)
        state.append(dict(part=part, new_id=new_id, ndata=ndata, g=g,
                          num_in=part.num_in, nt=nt, own=own))
    # boundary[i][j] = rank i's inner (renumbered) rows that rank j needs
    boundary = [[None] * world for _ in range(world)]
    pl = [[None] * world for _ in range(world)]
    for i in range(world):
        tot = state[i]["num_in"]
        for j in range(world):
            if j == i:
                continue
            sel = state[i]["own"] == j
            # what i needs FROM j, in j-local old ids (ascending)
            wanted = state[i]["part"].halo_gnid[sel] - offs[j]
            boundary[j][i] = state[j]["new_id"][wanted]
            pl[i][j] = (tot, tot + int(sel.sum()))
            tot += int(sel.sum())

    n_feat = parts[0].n_feat
    n_class = parts[0].n_class
    torch.manual_seed(seed)
    model = GraphSAGE([n_feat, hidden, n_class], F.relu, use_pp=False,
                      dropout=0.0, norm=None, n_linear=0)
    opt = torch.optim.Adam(model.parameters(), lr=lr)
    loss_fcn = torch.nn.CrossEntropyLoss(reduction="sum")

    L = 2  # conv layers
    sizes = [n_feat, hidden]
    stale_feat = [[{j: torch.zeros(boundary[j][i].numel(), sizes[l])
                    for j in range(world) if j != i}
                   for l in range(L)] for i in range(world)]
    stale_grad = [[{j: torch.zeros(boundary[i][j].numel(), sizes[l])
                    for j in range(world) if j != i}
                   for l in range(L)] for i in range(world)]
    avg_feat = [[{j: torch.zeros(boundary[j][i].numel(), sizes[l])
                  for j in range(world) if j != i}
                 for l in range(L)] for i in range(world)]
    avg_grad = [[{j: torch.zeros(boundary[i][j].numel(), sizes[l])
                  for j in range(world) if j != i}
                 for l in range(L)] for i in range(world)]
    # stale_feat[i][l][j]: values rank i RECEIVED from j (j's boundary rows)
    # stale_grad[i][l][j]: grads rank i RECEIVED from j (for i's boundary
    #                      rows listed in boundary[i][j])

    losses = []
    for epoch in range(n_epochs):
        next_feat = [[{} for _ in range(L)] for _ in range(world)]
        next_grad = [[{} for _ in range(L)] for _ in range(world)]
        total = torch.zeros(())
        opt.zero_grad(set_to_none=True)
        for i in range(world):
            st = state[i]
            h = st["ndata"]["feat"]
            inv = 1.0 / st["ndata"]["in_degree"].clamp(min=1)
            for l in range(L):
                # record what i sends this epoch (its boundary rows of h)
                for j in range(world):
                    if j != i:
                        next_feat[j][l][i] = h[boundary[i][j]].detach()
                src_f = avg_feat if corr else stale_feat
                h_all = torch.cat(
                    [h] + [src_f[i][l][j] for j in range(world)
                           if j != i])
                if l > 0 and h_all.requires_grad:
                    def make_hook(i=i, l=l):
                        def hook(grad):
                            g2 = grad.clone()
                            src_g = avg_grad if corr else stale_grad
                            for j in range(world):
                                if j == i:
                                    continue
                                # inject stale grads into my boundary rows
                                g2[boundary[i][j]] += src_g[i][l][j]
                                # record halo-slice grads to send to j
                                a, b = pl[i][j]
                                next_grad[j][l][i] = grad[a:b].detach()
                            return g2
                        return hook
                    h_all.register_hook(make_hook())
                layer = model.layers[l]
                from pipegcn_amd import ops
                ah = ops.spmm_mean(st["g"], h_all, inv.contiguous())
                h = layer.linear1(h_all[: st["num_in"]]) + layer.linear2(ah)
                if l < L - 1:
                    h = F.relu(h)
            loss = loss_fcn(h[: st["nt"]],
                            st["ndata"]["label"][: st["nt"]])
            total = total + loss
        total.backward()
        for p in model.parameters():
            p.grad /= n_train_global
        opt.step()
        losses.append(total.item())
        if corr:
            # the EMA updates when this epoch's transfer ARRIVES; it is
            # consumed next epoch
            for i in range(world):
                for l in range(L):
                    for j in range(world):
                        if j == i:
                            continue
                        avg_feat[i][l][j] = (momentum * avg_feat[i][l][j]
                                             + (1 - momentum)
                                             * next_feat[i][l][j])
                        ng = next_grad[i][l].get(
                            j, torch.zeros(boundary[i][j].numel(),
                                           sizes[l]))
                        avg_grad[i][l][j] = (momentum * avg_grad[i][l][j]
                                             + (1 - momentum) * ng)
        stale_feat, stale_grad = next_feat, next_grad
        # re-wrap dict layout: next_* entries keyed by sender
        stale_feat = [[{j: next_feat[i][l][j] for j in range(world)
                        if j != i} for l in range(L)] for i in range(world)]
        stale_grad = [[{j: next_grad[i][l].get(
            j, torch.zeros(boundary[i][j].numel(), sizes[l]))
            for j in range(world) if j != i} for l in range(L)]
            for i in range(world)]
    return losses


def _dist_worker(rank, world, tmpdir, corr=False):
    from pipegcn_amd import trainer
    from pipegcn_amd.graph.datasets import data_stats
    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.parallel.buffer import Buffer
    from pipegcn_amd.parallel.reducer import Reducer

    import torch.distributed as dist

    os.chdir(tmpdir)
    ctx.buffer = Buffer()
    ctx.reducer = Reducer()
    (u, v, n, ndata), part = _prepare_partitions(tmpdir + "/p", world)
    args = make_args(n_partitions=world, enable_pipeline=True, n_layers=2,
                     norm="none", n_epochs=5, lr=0.05, feat_corr=corr,
                     grad_corr=corr, corr_momentum=0.8)
    args.norm = None
    args.n_feat, args.n_class, args.n_train = data_stats(ndata)
    s = trainer.run(part, args, device="cpu")
    t = torch.tensor(s["losses"])
    dist.all_reduce(t)
    return t.tolist(), args.n_train


def test_pipelined_matches_sequential_simulator(tmp_path):
    from pipegcn_amd.graph import partition

    dist_losses, n_train = run_distributed(
        _dist_worker, WORLD, args=(str(tmp_path),))[0]

    parts = [partition.load_partition(str(tmp_path / "p"), r)
             for r in range(WORLD)]
    sim_losses = _simulate(parts, 5, hidden=16, lr=0.05,
                           n_train_global=n_train, seed=5)
    for a, b in zip(sim_losses, dist_losses):
        assert abs(a - b) / max(abs(a), 1e-9) < 2e-3, \
            (sim_losses, dist_losses)


def _dist_worker_corr(rank, world, tmpdir):
    return _dist_worker(rank, world, tmpdir, corr=True)


def test_pipelined_corrections_match_simulator(tmp_path):
    """Pipelined + feat/grad EMA correction vs the sequential simulator."""
    from pipegcn_amd.graph import partition

    dist_losses, n_train = run_distributed(
        _dist_worker_corr, WORLD, args=(str(tmp_path),))[0]
    parts = [partition.load_partition(str(tmp_path / "p"), r)
             for r in range(WORLD)]
    sim_losses = _simulate(parts, 5, hidden=16, lr=0.05,
                           n_train_global=n_train, seed=5, corr=True)
    for a, b in zip(sim_losses, dist_losses):
        assert abs(a - b) / max(abs(a), 1e-9) < 2e-3, \
            (sim_losses, dist_losses)
