"""Multi-process (gloo/CPU, world_size 2) protocol + equivalence tests.

These exercise the distributed runtime exactly as it runs on GPU (same code
path, RingTransport's gloo branch instead of RCCL): boundary exchange,
vanilla + pipelined Buffer semantics (epoch-0 zeros, one-epoch staleness,
EMA correction), gradient push-back, reducer averaging, SyncBatchNorm, and
the P-partition == 1-partition training equivalence.
"""
import os
import types

import torch
import torch.distributed as dist

from pipegcn_amd.utils.timer import comm_timer
from tests.conftest import run_distributed

WORLD = 2


def make_args(**kw):
    base = dict(dataset="synth-small", model="graphsage", dropout=0.0,
                lr=0.01, n_epochs=6, n_partitions=WORLD, n_hidden=16,
                n_layers=3, n_linear=0, norm="layer", weight_decay=0,
                n_feat=0, n_class=0, n_train=0, enable_pipeline=False,
                feat_corr=False, grad_corr=False, corr_momentum=0.95,
                use_pp=False, inductive=False, seed=5, log_every=100,
                backend="gloo", eval=False, graph_name="test",
                dtype="fp32")
    base.update(kw)
    return types.SimpleNamespace(**base)


def _prepare_partitions(tmpdir, nparts, shape="small", seed=5):
    """Rank 0 partitions the global graph; all ranks load their part.

    nparts_hint is pinned (NOT nparts) so different partition counts see the
    IDENTICAL global graph — required by the equivalence test.
    """
    from pipegcn_amd.graph import partition, synthetic

    u, v, n, ndata = synthetic.synth_global(shape, nparts_hint=2, seed=seed)
    if dist.get_rank() == 0:
        partition.partition_and_save(u, v, n, ndata, tmpdir, nparts,
                                     "metis", "vol", 0)
    dist.barrier()
    part = partition.load_partition(tmpdir, dist.get_rank())
    return (u, v, n, ndata), part


# ---------------------------------------------------------------- halo agg


def _halo_agg_worker(rank, world, tmpdir):
    from pipegcn_amd import ops
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    (u, v, n, ndata), part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)

    # distributed mean aggregation of the raw features through the Buffer
    buf = Buffer()
    F = rp.ndata["feat"].shape[1]
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F], pipeline=False, backend="gloo")
    h_all = buf.update(0, rp.ndata["feat"])
    inv_deg = 1.0 / rp.ndata["in_degree"].clamp(min=1)
    ah = ops.spmm_mean(rp.graph, h_all, inv_deg)
    buf.shutdown()

    # reference: global mean aggregation (recomputed identically everywhere)
    from pipegcn_amd.graph.partition import assign_partitions
    pvec = assign_partitions(u, v, n, world, "metis", "vol", 0).long()
    order = torch.argsort(pvec, stable=True)
    new_gid = torch.empty(n, dtype=torch.long)
    new_gid[order] = torch.arange(n)
    gfeat = torch.zeros(n, F)
    gfeat[new_gid] = ndata["feat"]
    deg = torch.bincount(new_gid[v], minlength=n).float().clamp(min=1)
    ref = torch.zeros(n, F)
    ref.index_add_(0, new_gid[v], gfeat[new_gid[u]])
    ref /= deg.unsqueeze(1)

    gid = rp.ndata["gid"]
    assert torch.allclose(ah, ref[gid], atol=1e-4), \
        (ah - ref[gid]).abs().max()
    return True


def test_halo_aggregation_matches_full_graph(tmp_path):
    run_distributed(_halo_agg_worker, WORLD, args=(str(tmp_path),))


# ----------------------------------------------------------- pipeline mode


def _pipeline_worker(rank, world, tmpdir):
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    _, part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)
    F = 4
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F], pipeline=True, backend="gloo")
    peer = 1 - rank
    nhalo = rp.recv_shape[peer]
    for epoch in range(3):
        feat = torch.full((rp.num_in, F), float(epoch + 1 + 10 * rank))
        h = buf.update(0, feat)
        halo_part = h[rp.num_in:rp.num_in + nhalo]
        if epoch == 0:
            assert torch.equal(halo_part, torch.zeros(nhalo, F)), \
                "epoch 0 must consume zeros"
        else:
            expect = float(epoch + 10 * peer)  # previous epoch's value
            assert torch.equal(halo_part,
                               torch.full((nhalo, F), expect)), \
                f"epoch {epoch}: staleness must be exactly 1"
        buf.next_epoch()
        comm_timer.clear()
    buf.synchronize()
    buf.shutdown()
    return True


def test_pipeline_one_epoch_staleness(tmp_path):
    run_distributed(_pipeline_worker, WORLD, args=(str(tmp_path),))


def _corr_worker(rank, world, tmpdir):
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    _, part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)
    F, m = 3, 0.5
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F], pipeline=True, backend="gloo", corr_feat=True,
                    corr_momentum=m)
    peer = 1 - rank
    nhalo = rp.recv_shape[peer]
    avg = 0.0
    for epoch in range(4):
        feat = torch.full((rp.num_in, F), float(epoch + 1 + 10 * rank))
        h = buf.update(0, feat)
        halo_part = h[rp.num_in:rp.num_in + nhalo]
        assert torch.allclose(halo_part, torch.full((nhalo, F), avg),
                              atol=1e-5), f"epoch {epoch}"
        # the EMA lands after this epoch's transfer (peer's current value)
        avg = m * avg + (1 - m) * float(epoch + 1 + 10 * peer)
        buf.next_epoch()
        comm_timer.clear()
    buf.synchronize()
    buf.shutdown()
    return True


def test_pipeline_feat_correction_ema(tmp_path):
    run_distributed(_corr_worker, WORLD, args=(str(tmp_path),))


# ------------------------------------------------------------- grad push


def _grad_worker(rank, world, tmpdir):
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    _, part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)
    F = 4
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F, F], pipeline=False, backend="gloo")
    feat = torch.randn(rp.num_in, F).requires_grad_(True)
    h = buf.update(1, feat)  # layer 1 => grads flow
    h.sum().backward()
    # expected: ones + one extra per peer that lists the node as boundary
    expect = torch.ones(rp.num_in, F)
    for j in range(world):
        if j != rank and rp.boundary[j] is not None:
            expect[rp.boundary[j]] += 1.0
    assert torch.allclose(feat.grad, expect), \
        (feat.grad - expect).abs().max()
    buf.shutdown()
    return True


def test_vanilla_grad_pushback(tmp_path):
    run_distributed(_grad_worker, WORLD, args=(str(tmp_path),))


# ------------------------------------------------------------- reducer


def _reducer_worker(rank, world):
    from pipegcn_amd.parallel.reducer import Reducer

    torch.manual_seed(0)
    model = torch.nn.Linear(4, 3)
    red = Reducer()
    red.init(model)
    for p in model.parameters():
        p.grad = torch.full_like(p, float(rank + 1))
    n_train = 10
    red.synchronize(n_train)
    want = (1.0 + 2.0) / n_train
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p, want))
    return True


def test_reducer_global_mean():
    run_distributed(_reducer_worker, WORLD)


# ------------------------------------------------------------- sync BN


def _bn_worker(rank, world):
    from pipegcn_amd.models.sync_bn import SyncBatchNorm

    torch.manual_seed(42)
    xs = [torch.randn(50 + 10 * r, 8) for r in range(world)]
    x = xs[rank].clone().requires_grad_(True)
    whole = sum(t.shape[0] for t in xs)
    bn = SyncBatchNorm(8, whole)
    y = bn(x)
    gout = torch.ones_like(y) * torch.arange(8)
    y.backward(gout)

    # single-process reference over the concatenated rows
    xa = torch.cat(xs).requires_grad_(True)
    mean = xa.mean(0)
    var = (xa * xa).mean(0) - mean * mean
    xh = (xa - mean.detach()) / torch.sqrt(var.detach() + 1e-5)
    # reference semantics: mean/var treated as constants in backward
    n0 = sum(t.shape[0] for t in xs[:rank])
    ya = xh * 1.0 + 0.0
    assert torch.allclose(y.detach(), ya[n0:n0 + x.shape[0]].detach(),
                          atol=1e-4)
    # backward formula check: dx = (w/n)/std*(n*g - sum g - xh * sum(g*xh))
    g_all = torch.ones(whole, 8) * torch.arange(8)
    std = torch.sqrt(var.detach() + 1e-5)
    dbias = g_all.sum(0)
    dweight = (g_all * xh.detach()).sum(0)
    dx = (1.0 / whole) / std * (whole * g_all - dbias
                                - xh.detach() * dweight)
    assert torch.allclose(x.grad, dx[n0:n0 + x.shape[0]], atol=1e-4)
    return True


def test_sync_batchnorm():
    run_distributed(_bn_worker, WORLD)


# -------------------------------------------------- training equivalence


def _equiv_worker(rank, world, tmpdir, pipeline, model="graphsage",
                  dtype="fp32"):
    from pipegcn_amd import trainer
    from pipegcn_amd.graph import partition
    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.parallel.buffer import Buffer
    from pipegcn_amd.parallel.reducer import Reducer

    os.chdir(tmpdir)
    ctx.buffer = Buffer()
    ctx.reducer = Reducer()
    (u, v, n, ndata), part = _prepare_partitions(
        os.path.join(tmpdir, f"p{world}"), world)
    args = make_args(n_partitions=world,
                     enable_pipeline=pipeline, model=model, dtype=dtype)
    from pipegcn_amd.graph.datasets import data_stats
    args.n_feat, args.n_class, args.n_train = data_stats(ndata)
    s = trainer.run(part, args, device="cpu")
    # global loss per epoch = sum over ranks
    t = torch.tensor(s["losses"])
    dist.all_reduce(t)
    return t.tolist()


def _single_worker(rank, world, tmpdir):
    return _equiv_worker(rank, world, tmpdir, False)


def test_2part_matches_1part_training(tmp_path):
    """P=2 vanilla (non-pipelined) training must match P=1 (same seed,
    dropout 0) — reference test strategy."""
    two = run_distributed(_single_worker, 2, args=(str(tmp_path),))[0]
    one = run_distributed(_single_worker, 1, args=(str(tmp_path),))[0]
    for a, b in zip(one, two):
        assert abs(a - b) / max(abs(a), 1e-9) < 2e-3, (one, two)


def test_pipelined_training_converges(tmp_path):
    """Pipelined (stale) training: loss decreases over epochs."""
    import functools

    losses = run_distributed(
        functools.partial(_equiv_worker, pipeline=True), 2,
        args=(str(tmp_path),))[0]
    assert losses[-1] < losses[0]


def _gcn_worker(rank, world, tmpdir):
    return _equiv_worker(rank, world, tmpdir, False, model="gcn")


def test_gcn_2part_matches_1part(tmp_path):
    two = run_distributed(_gcn_worker, 2, args=(str(tmp_path),))[0]
    one = run_distributed(_gcn_worker, 1, args=(str(tmp_path),))[0]
    for a, b in zip(one, two):
        assert abs(a - b) / max(abs(a), 1e-9) < 2e-3, (one, two)
    assert two[-1] < two[0]


# --------------------------------------------- disconnected partition pair


def _chain_worker(rank, world, tmpdir):
    """3 partitions in a chain 0-1-2: ranks 0 and 2 share NO edges — their
    boundary/recv sets are empty and the batched P2P exchange must skip them
    consistently on both sides (RCCL hazard at 8 GPUs)."""
    from pipegcn_amd.graph import partition
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    if rank == 0:
        # chain graph: 3 blocks of 10 nodes; edges only within a block or
        # to the adjacent block; plus self-loops
        u, v = [], []
        n = 30
        for i in range(n):
            u.append(i)
            v.append(i)
            blk = i // 10
            for j in range(3):
                t = (i + j * 3 + 1) % 10 + blk * 10
                u.append(t)
                v.append(i)
            if blk < 2:  # edge from next block
                u.append(i + 10)
                v.append(i)
        # force block partition by pre-assigning: use random method w/ seed
        # -> instead use explicit partition via contiguous split: the
        # partitioner with balanced blocks on a chain finds blocks anyway
        partition.partition_and_save(torch.tensor(u), torch.tensor(v), n,
                                     {"feat": torch.randn(n, 4),
                                      "label": torch.randint(0, 3, (n,)),
                                      "train_mask": torch.ones(n).bool(),
                                      "val_mask": torch.zeros(n).bool(),
                                      "test_mask": torch.zeros(n).bool()},
                                     tmpdir, 3, "metis", "cut", 1)
    dist.barrier()
    part = partition.load_partition(tmpdir, rank)
    rp = build_runtime_partition(part)
    # sanity: SOME pair must be empty for the test to mean anything;
    # find it dynamically (partitioner may order blocks arbitrarily)
    empties = sum(1 for j in range(world)
                  if j != rank and rp.recv_shape[j] == 0)
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [4, 4], pipeline=True, backend="gloo")
    for epoch in range(3):
        feat = torch.full((rp.num_in, 4), float(epoch + 1),
                          requires_grad=True)
        h = buf.update(1, feat)
        h.sum().backward()
        buf.next_epoch()
        comm_timer.clear()
    buf.synchronize()
    buf.shutdown()
    return empties


def test_chain_partition_empty_boundary(tmp_path):
    empties = run_distributed(_chain_worker, 3, args=(str(tmp_path),))
    # the chain's two end partitions each see one empty peer
    assert sum(empties) >= 2, empties


def _bf16_worker(rank, world, tmpdir):
    return _equiv_worker(rank, world, tmpdir, True, model="graphsage",
                         dtype="bf16")


def test_bf16_pipelined_training_converges(tmp_path):
    """Optional bf16 compute: pipelined 2-rank training converges and the
    bf16 losses track the fp32 run to a few percent."""
    import functools

    bf = run_distributed(_bf16_worker, 2, args=(str(tmp_path),))[0]
    fp = run_distributed(
        functools.partial(_equiv_worker, pipeline=True), 2,
        args=(str(tmp_path),))[0]
    assert bf[-1] < bf[0]
    for a, b in zip(fp, bf):
        assert abs(a - b) / abs(a) < 0.05, (fp, bf)


def _vanilla_corr_worker(rank, world, tmpdir):
    """Corrections in NON-pipelined mode: the EMA tracks the FRESH value of
    each epoch (no staleness), consumed instead of the raw recv."""
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer

    _, part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)
    F, m = 3, 0.5
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F], pipeline=False, backend="gloo", corr_feat=True,
                    corr_momentum=m)
    peer = 1 - rank
    nhalo = rp.recv_shape[peer]
    avg = 0.0
    for epoch in range(3):
        feat = torch.full((rp.num_in, F), float(epoch + 1 + 10 * rank))
        h = buf.update(0, feat)
        # synchronous mode: this epoch's value is already in the EMA
        avg = m * avg + (1 - m) * float(epoch + 1 + 10 * peer)
        halo = h[rp.num_in:rp.num_in + nhalo]
        assert torch.allclose(halo, torch.full((nhalo, F), avg),
                              atol=1e-5), f"epoch {epoch}"
        buf.next_epoch()
        comm_timer.clear()
    buf.shutdown()
    return True


def test_vanilla_mode_feat_correction(tmp_path):
    run_distributed(_vanilla_corr_worker, WORLD, args=(str(tmp_path),))


def _pp_worker(rank, world, tmpdir):
    """--use-pp precompute: [feat ‖ mean_agg] must equal the global-graph
    mean aggregation (reference train.py:169-189)."""
    import types

    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.trainer import precompute

    (u, v, n, ndata), part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part)
    feat_pp = precompute(rp, types.SimpleNamespace(model="graphsage"))
    F = rp.ndata["feat"].shape[1]
    assert feat_pp.shape == (rp.num_in, 2 * F)

    # global reference (same mapping as the halo-agg test)
    from pipegcn_amd.graph.partition import assign_partitions
    pvec = assign_partitions(u, v, n, world, "metis", "vol", 0).long()
    order = torch.argsort(pvec, stable=True)
    new_gid = torch.empty(n, dtype=torch.long)
    new_gid[order] = torch.arange(n)
    gfeat = torch.zeros(n, F)
    gfeat[new_gid] = ndata["feat"]
    deg = torch.bincount(new_gid[v], minlength=n).float().clamp(min=1)
    ref = torch.zeros(n, F)
    ref.index_add_(0, new_gid[v], gfeat[new_gid[u]])
    ref /= deg.unsqueeze(1)
    gid = rp.ndata["gid"]
    assert torch.allclose(feat_pp[:, :F], rp.ndata["feat"], atol=1e-6)
    assert torch.allclose(feat_pp[:, F:], ref[gid], atol=1e-4), \
        (feat_pp[:, F:] - ref[gid]).abs().max()
    return True


def test_use_pp_precompute_matches_global(tmp_path):
    run_distributed(_pp_worker, WORLD, args=(str(tmp_path),))
