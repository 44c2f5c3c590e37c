"""Per-process training driver (reference: /root/reference/train.py:242-416).

Flow: load partition -> build halo graph (boundary ring exchange) -> init
pipelined Buffer -> optional --use-pp precompute -> model + Reducer ->
epoch loop (forward / sum-loss / backward / next_epoch / reducer.synchronize /
Adam step) -> async rank-0 CPU full-graph eval every --log-every epochs ->
save best state_dict to model/<graph_name>_final.pth.tar (reference format,
train.py:397; we also create the directory — the reference forgot to,
train.py:258-259).
"""
from __future__ import annotations

import copy
import os
import time
from multiprocessing.pool import ThreadPool
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist
import torch.nn.functional as F

from pipegcn_amd import ops
from pipegcn_amd.cli import check_args
from pipegcn_amd.graph import datasets
from pipegcn_amd.graph.csr import FullGraph
from pipegcn_amd.graph.halo import RuntimePartition, build_runtime_partition
from pipegcn_amd.graph.partition import PartData, load_partition
from pipegcn_amd.models.sage import GraphSAGE
from pipegcn_amd.parallel import context as ctx
from pipegcn_amd.utils.timer import comm_timer


def get_layer_size(n_feat, n_hidden, n_class, n_layers):
    return [n_feat] + [n_hidden] * (n_layers - 1) + [n_class]


def calc_acc(logits, labels):
    if labels.dim() == 1:
        indices = logits.argmax(dim=1)
        return (indices == labels).sum().item() / labels.shape[0]
    from sklearn.metrics import f1_score

    return f1_score(labels, logits > 0, average="micro")


@torch.no_grad()
def evaluate_trans(name, model, g: FullGraph, result_file_name=None):
    model.eval()
    model.cpu()
    feat, labels = g.ndata["feat"], g.ndata["label"]
    feat = feat.to(next(model.parameters()).dtype)
    logits = model(g, feat).float()
    val_acc = calc_acc(logits[g.ndata["val_mask"]],
                       labels[g.ndata["val_mask"]])
    test_acc = calc_acc(logits[g.ndata["test_mask"]],
                        labels[g.ndata["test_mask"]])
    buf = "{:s} | Validation Accuracy {:.2%} | Test Accuracy {:.2%}".format(
        name, val_acc, test_acc)
    if result_file_name is not None:
        with open(result_file_name, "a+") as f:
            f.write(buf + "\n")
    print(buf)
    return model, val_acc


@torch.no_grad()
def evaluate_induc(name, model, g: FullGraph, mode, result_file_name=None):
    """mode: 'val' or 'test'"""
    model.eval()
    model.cpu()
    feat, labels = g.ndata["feat"], g.ndata["label"]
    feat = feat.to(next(model.parameters()).dtype)
    mask = g.ndata[mode + "_mask"]
    logits = model(g, feat).float()
    acc = calc_acc(logits[mask], labels[mask])
    buf = "{:s} | Accuracy {:.2%}".format(name, acc)
    if result_file_name is not None:
        with open(result_file_name, "a+") as f:
            f.write(buf + "\n")
    print(buf)
    return model, acc


def node_subgraph(u, v, n, ndata, mask):
    """Induced subgraph on mask (reference uses g.subgraph)."""
    idx = mask.nonzero(as_tuple=True)[0]
    new_id = torch.full((n,), -1, dtype=torch.long)
    new_id[idx] = torch.arange(idx.numel())
    keep = mask[u] & mask[v]
    su, sv = new_id[u[keep]], new_id[v[keep]]
    sub_ndata = {k: t[idx] for k, t in ndata.items()}
    return su, sv, idx.numel(), sub_ndata


def inductive_split(u, v, n, ndata):
    """(reference /root/reference/helper/utils.py:226-230)"""
    tr = node_subgraph(u, v, n, ndata, ndata["train_mask"])
    va = node_subgraph(u, v, n, ndata,
                       ndata["train_mask"] | ndata["val_mask"])
    return tr, va, (u, v, n, ndata)


def exchange_halo_values(rp: RuntimePartition,
                         values: torch.Tensor) -> torch.Tensor:
    """One-shot fetch of a per-node scalar (e.g. in-degree) for halo nodes;
    returns values extended to [num_all] in halo-slot order."""
    size = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    if size == 1:
        return _pad_halo_zeros(rp, values.unsqueeze(1)).squeeze(1)
    from pipegcn_amd.parallel.transport import RingTransport

    send = [None] * size
    recv = [None] * size
    col = values.unsqueeze(1)
    for j in range(size):
        if j == rank:
            continue
        send[j] = col[rp.boundary[j]].contiguous()
        recv[j] = torch.zeros(rp.recv_shape[j], 1, device=values.device)
    RingTransport().all_to_all(send, recv, key="halo_scalar", tag=2)
    return torch.cat([values] + [recv[j].squeeze(1) for j in range(size)
                                 if j != rank])


def _pad_halo_zeros(rp: RuntimePartition, t: torch.Tensor) -> torch.Tensor:
    """Extend [num_in, F] to [num_all, F] with zero halo rows (solo sizing
    mode has real halo slots but no peers to fill them — epoch-0 semantics)."""
    if t.shape[0] >= rp.num_all:
        return t
    pad = torch.zeros(rp.num_all - t.shape[0], *t.shape[1:],
                      device=t.device, dtype=t.dtype)
    return torch.cat([t, pad])


def precompute(rp: RuntimePartition, args) -> torch.Tensor:
    """--use-pp: one-shot raw-feature halo exchange + one mean-agg SpMM;
    returns [feat ‖ mean_feat] so layer 0 needs no per-epoch communication
    (reference /root/reference/train.py:169-189)."""
    if args.model != "graphsage":
        raise NotImplementedError(
            "--use-pp supports graphsage only (reference parity)")
    feat = rp.ndata["feat"]
    size = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    if size > 1:
        from pipegcn_amd.parallel.transport import RingTransport

        send = [None] * size
        recv = [None] * size
        for j in range(size):
            if j == rank:
                continue
            send[j] = feat[rp.boundary[j]].contiguous()
            recv[j] = torch.zeros(rp.recv_shape[j], feat.shape[1],
                                  device=feat.device)
        RingTransport().all_to_all(send, recv, key="pp", tag=1)
        feat_all = torch.cat([feat] + [recv[j] for j in range(size)
                                       if j != rank])
    else:
        # solo sizing mode: the halo graph references [0, num_all) source
        # rows even with no peers — pad with zeros (an under-sized feat
        # tensor would be an out-of-bounds SpMM gather)
        feat_all = _pad_halo_zeros(rp, feat)
    inv_deg = (1.0 / rp.ndata["in_degree"].clamp(min=1.0)).contiguous()
    mean_feat = ops.spmm(rp.graph.csr, feat_all, inv_deg)
    return torch.cat([feat, mean_feat], dim=1)


def create_model(layer_size, args):
    if args.model == "graphsage":
        return GraphSAGE(layer_size, F.relu, args.use_pp, norm=args.norm,
                         dropout=args.dropout, n_linear=args.n_linear,
                         train_size=args.n_train)
    if args.model == "gcn":
        from pipegcn_amd.models.gcn import GCN

        return GCN(layer_size, F.relu, args.use_pp, norm=args.norm,
                   dropout=args.dropout, n_linear=args.n_linear,
                   train_size=args.n_train)
    raise NotImplementedError(f"unknown model {args.model}")


def run(part: PartData, args, device: str = "cpu",
        eval_graphs=None) -> dict:
    """Train this rank's partition. Returns summary stats (rank-local)."""
    rank = dist.get_rank() if dist.is_initialized() else 0
    size = dist.get_world_size() if dist.is_initialized() else 1

    if rank == 0:
        os.makedirs("checkpoint/", exist_ok=True)
        os.makedirs("results/", exist_ok=True)
        os.makedirs("model/", exist_ok=True)

    val_g = test_g = None
    if rank == 0 and args.eval:
        if eval_graphs is None:
            u, v, n, ndata = datasets.load_data(
                args.dataset, nparts_hint=args.n_partitions, seed=args.seed)
            if args.inductive:
                _, (vu, vv, vn, vnd), (tu, tv, tn, tnd) = inductive_split(
                    u, v, n, ndata)
                val_g = FullGraph.from_coo(vu, vv, vn, vnd)
                test_g = FullGraph.from_coo(tu, tv, tn, tnd)
            else:
                val_g = test_g = FullGraph.from_coo(u, v, n, ndata)
        else:
            val_g, test_g = eval_graphs

    rp = build_runtime_partition(part, device=device)
    print(f"Process {rank} has {rp.num_all} nodes, {rp.graph.csr.nnz} edges, "
          f"{rp.num_in} inner nodes ({rp.num_train} train).")

    layer_size = get_layer_size(args.n_feat, args.n_hidden, args.n_class,
                                args.n_layers)

    comm_group = dist.new_group(backend=args.backend) if size > 1 else None
    ctx.buffer.init_buffer(
        rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
        layer_size[: args.n_layers - args.n_linear], use_pp=args.use_pp,
        backend=args.backend, pipeline=args.enable_pipeline,
        corr_feat=args.feat_corr, corr_grad=args.grad_corr,
        corr_momentum=args.corr_momentum, device=device, group=comm_group,
        collect_stats=getattr(args, "collect_stats", False),
        dtype=(torch.bfloat16 if getattr(args, "dtype", "fp32") == "bf16"
               else torch.float32))

    dtype = (torch.bfloat16 if getattr(args, "dtype", "fp32") == "bf16"
             else torch.float32)
    if dtype != torch.float32:
        rp.ndata["feat"] = rp.ndata["feat"].to(dtype)
    feat = rp.ndata["feat"]
    if args.use_pp:
        feat = precompute(rp, args)

    in_deg = rp.ndata["in_degree"]
    if args.model == "gcn":
        # GCN's symmetric normalization needs halo-node degrees too
        in_deg = exchange_halo_values(rp, in_deg)
    labels_all = rp.ndata["label"]
    num_train = rp.num_train
    part_train = max(num_train, 1)
    labels = labels_all[:num_train]

    torch.manual_seed(args.seed)
    model = create_model(layer_size, args)
    model = model.to(device)
    if dtype != torch.float32:
        model = model.to(dtype)

    ctx.reducer.init(model)

    if args.dataset == "yelp" or labels.dim() > 1:
        loss_fcn = torch.nn.BCEWithLogitsLoss(reduction="sum")
    else:
        loss_fcn = torch.nn.CrossEntropyLoss(reduction="sum")
    optimizer = torch.optim.Adam(model.parameters(), lr=args.lr,
                                 weight_decay=args.weight_decay)

    if args.grad_corr and args.feat_corr:
        suffix = "_grad_feat"
    elif args.grad_corr:
        suffix = "_grad"
    elif args.feat_corr:
        suffix = "_feat"
    else:
        suffix = ""
    result_file_name = "results/%s_n%d_p%d%s.txt" % (
        args.dataset, args.n_partitions, int(args.enable_pipeline), suffix)

    train_dur, comm_dur, reduce_dur, loss_hist = [], [], [], []
    best_model, best_acc = None, 0.0
    thread = None
    pool = ThreadPool(processes=1)
    if device != "cpu":
        torch.cuda.reset_peak_memory_stats()

    # --- checkpoint/resume (extension; reference has save-only final model)
    ckpt_every = getattr(args, "checkpoint_every", 0)
    ckpt_path = os.path.join(
        "checkpoint", f"{args.graph_name}_rank{rank}.ckpt")
    start_epoch = 0
    if getattr(args, "resume", False) and os.path.exists(ckpt_path):
        state = torch.load(ckpt_path, map_location=device,
                           weights_only=True)
        model.load_state_dict(state["model"])
        optimizer.load_state_dict(state["optimizer"])
        start_epoch = state["epoch"] + 1
        best_acc = state.get("best_acc", 0.0)
        print(f"Process {rank:03d} | resumed from epoch {state['epoch']}")

    for epoch in range(start_epoch, args.n_epochs):
        t0 = time.time()
        model.train()
        logits = model(rp.graph, feat, in_deg)
        if logits.dtype != torch.float32:
            logits = logits.float()  # loss in fp32
        if args.inductive:
            loss = loss_fcn(logits, labels_all)
        else:
            loss = loss_fcn(logits[:num_train], labels)
        del logits
        optimizer.zero_grad(set_to_none=True)
        loss.backward()

        ctx.buffer.next_epoch()
        pre_reduce = time.time()
        ctx.reducer.synchronize(args.n_train)
        reduce_time = time.time() - pre_reduce
        optimizer.step()

        if epoch >= 5 and epoch % args.log_every != 0:
            train_dur.append(time.time() - t0)
            comm_dur.append(comm_timer.tot_time())
            reduce_dur.append(reduce_time)

        if (epoch + 1) % 10 == 0:
            print("Process {:03d} | Epoch {:05d} | Time(s) {:.4f} | "
                  "Comm(s) {:.4f} | Reduce(s) {:.4f} | Loss {:.4f}".format(
                      rank, epoch, float(np.mean(train_dur or [0])),
                      float(np.mean(comm_dur or [0])),
                      float(np.mean(reduce_dur or [0])),
                      loss.item() / part_train))
        comm_timer.clear()
        loss_hist.append(loss.item())
        del loss

        if ckpt_every > 0 and (epoch + 1) % ckpt_every == 0:
            os.makedirs("checkpoint/", exist_ok=True)
            torch.save({"epoch": epoch, "model": model.state_dict(),
                        "optimizer": optimizer.state_dict(),
                        "best_acc": best_acc}, ckpt_path)

        if (rank == 0 and args.eval
                and (epoch + 1) % args.log_every == 0):
            if thread is not None:
                model_copy, val_acc = thread.get()
                if val_acc > best_acc:
                    best_acc, best_model = val_acc, model_copy
            model_copy = copy.deepcopy(model)
            if not args.inductive:
                thread = pool.apply_async(
                    evaluate_trans,
                    args=("Epoch %05d" % epoch, model_copy, val_g,
                          result_file_name))
            else:
                thread = pool.apply_async(
                    evaluate_induc,
                    args=("Epoch %05d" % epoch, model_copy, val_g, "val",
                          result_file_name))

    ctx.buffer.synchronize()
    if device != "cpu":
        print("Process {:03d} | peak GPU memory {:.2f} GB".format(
            rank, torch.cuda.max_memory_allocated() / 2**30))
    summary = {
        "rank": rank,
        "losses": loss_hist,
        "mean_epoch_s": float(np.mean(train_dur)) if train_dur else None,
        "mean_comm_s": float(np.mean(comm_dur)) if comm_dur else None,
        "mean_reduce_s": float(np.mean(reduce_dur)) if reduce_dur else None,
    }
    if args.eval and rank == 0:
        if thread is not None:
            model_copy, val_acc = thread.get()
            if val_acc > best_acc:
                best_acc, best_model = val_acc, model_copy
        if best_model is None:
            best_model = copy.deepcopy(model).cpu()
        torch.save(best_model.state_dict(),
                   "model/" + args.graph_name + "_final.pth.tar")
        print("model saved")
        print("Validation accuracy {:.2%}".format(best_acc))
        _, test_acc = evaluate_induc("Test Result", best_model, test_g,
                                     "test")
        summary["val_acc"] = best_acc
        summary["test_acc"] = test_acc
    ctx.buffer.shutdown()
    return summary


def init_processes(rank, size, args, device: Optional[str] = None):
    """Entry for spawned per-partition processes (reference train.py:408)."""
    os.environ["MASTER_ADDR"] = args.master_addr
    os.environ["MASTER_PORT"] = "%d" % args.port
    from datetime import timedelta

    dist.init_process_group(
        args.backend, rank=rank, world_size=size,
        timeout=timedelta(seconds=getattr(args, "dist_timeout", 1800)))
    check_args(args)
    if device is None:
        device = "cuda:0" if (args.backend == "nccl"
                              and torch.cuda.is_available()) else "cpu"
    if args.backend == "nccl" and not torch.cuda.is_available():
        raise RuntimeError(
            "--backend nccl (RCCL) needs a GPU; use --backend gloo on CPU")
    if device.startswith("cuda"):
        torch.cuda.set_device(torch.device(device))
    graph_dir = os.path.join("partitions", args.graph_name)
    part = load_partition(graph_dir, rank)
    try:
        return run(part, args, device=device)
    finally:
        dist.destroy_process_group()
