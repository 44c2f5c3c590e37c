"""Two ranks sharing ONE GPU (gloo backend, CUDA tensors).

Exercises the real GPU-side Buffer choreography — side comm stream, HIP
events, gather/scatter kernels, pinned staging — with world_size 2 on a
single MI355X (RCCL itself needs one GPU per rank, so the gloo+CUDA
transport branch stands in; the batched op-ordering branch is covered by the
CPU tier, and the full RCCL path by the driver's multi-GPU bench).
"""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu

WORLD = 2


def _pipeline_gpu_worker(rank, world, tmpdir):
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer
    from pipegcn_amd.utils.timer import comm_timer
    from tests.test_distributed import _prepare_partitions

    _, part = _prepare_partitions(tmpdir, world)
    rp = build_runtime_partition(part, device="cuda:0")
    F = 8
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F, F], pipeline=True, backend="gloo", device="cuda:0",
                    corr_feat=True, corr_momentum=0.5)
    peer = 1 - rank
    nhalo = rp.recv_shape[peer]
    avg = 0.0
    for epoch in range(4):
        # requires_grad from epoch 0 on: the grad hook must post a transfer
        # EVERY epoch (1-epoch staleness), exactly like real training
        feat = torch.full((rp.num_in, F), float(epoch + 1 + 10 * rank),
                          device="cuda:0", requires_grad=True)
        h = buf.update(1, feat)
        halo = h[rp.num_in:rp.num_in + nhalo]
        assert torch.allclose(halo.detach().cpu(),
                              torch.full((nhalo, F), avg), atol=1e-5), \
            f"epoch {epoch}: {halo[0,0].item()} != {avg}"
        h.sum().backward()
        assert torch.isfinite(feat.grad).all()
        avg = 0.5 * avg + 0.5 * float(epoch + 1 + 10 * peer)
        buf.next_epoch()
        comm_timer.clear()
    buf.synchronize()
    buf.shutdown()
    return True


def test_gpu_pipeline_two_ranks_one_device(tmp_path):
    run_distributed(_pipeline_gpu_worker, WORLD, args=(str(tmp_path),), timeout=900)


def _train_gpu_worker(rank, world, tmpdir):
    import os

    from pipegcn_amd import trainer
    from pipegcn_amd.graph.datasets import data_stats
    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.parallel.buffer import Buffer
    from pipegcn_amd.parallel.reducer import Reducer
    from tests.test_distributed import _prepare_partitions, make_args

    os.chdir(tmpdir)
    ctx.buffer = Buffer()
    ctx.reducer = Reducer()
    (u, v, n, ndata), part = _prepare_partitions(tmpdir + "/p", world)
    args = make_args(n_partitions=world, enable_pipeline=True, n_epochs=8)
    args.n_feat, args.n_class, args.n_train = data_stats(ndata)
    s = trainer.run(part, args, device="cuda:0")
    assert all(map(lambda x: x == x, s["losses"])), "NaN loss"
    assert s["losses"][-1] < s["losses"][0]
    return True


def test_gpu_two_rank_training_one_device(tmp_path):
    run_distributed(_train_gpu_worker, WORLD, args=(str(tmp_path),), timeout=900)
