"""Flagship benchmark: full-graph GraphSAGE training epoch time on a
synthetic Reddit-shaped graph (BASELINE.json metric: "epoch time (s) +
boundary-comm overlap %, Reddit 3-layer GraphSAGE at 1/2/4/8 parts").

One step == one full-graph training epoch (forward + sum-loss + backward +
pipelined boundary exchange + gradient all-reduce + Adam step), strong
scaling: the Reddit-shaped graph is partitioned over N GPUs.

Launch:  python bench.py --gpus N --steps K --warmup W
  N == 1: runs standalone.
  N > 1 : launched by the driver as
          python -m torch.distributed.run --nnodes=1 --nproc-per-node N
              --master-addr 127.0.0.1 bench.py --gpus N ...
          (reads RANK/WORLD_SIZE/MASTER_* from the env, RCCL over xGMI).

Rank 0 prints ONE JSON line with the whole-job metric (max epoch time over
ranks) plus the comm-overlap statistics measured with HIP events on the side
comm stream.
"""
import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--shape", type=str, default="reddit",
                    help="synthetic shape (reddit/ogbn-products/yelp/small)")
    ap.add_argument("--model", choices=["graphsage", "gcn"],
                    default="graphsage")
    ap.add_argument("--n-hidden", type=int, default=256)
    ap.add_argument("--n-layers", type=int, default=3)
    ap.add_argument("--no-pipeline", action="store_true",
                    help="disable cross-epoch pipelining (vanilla mode)")
    ap.add_argument("--use-pp", action="store_true",
                    help="precompute layer-0 aggregation (reference "
                         "headline config)")
    ap.add_argument("--norm", type=str, default="layer",
                    choices=["layer", "batch", "none"])
    ap.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32",
                    help="optional bf16 compute (the judged baseline "
                         "config is fp32 = reference parity)")
    ap.add_argument("--feat-corr", action="store_true")
    ap.add_argument("--grad-corr", action="store_true")
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--backend", type=str, default=None,
                    help="override torch.distributed backend (tests)")
    ap.add_argument("--solo-of", type=int, default=0,
                    help="sizing mode: run rank 0 of an N-way partition "
                         "alone on one GPU (full buffer layout + event "
                         "choreography, empty wire) — the papers100M "
                         "288 GB/GPU check: "
                         "`--shape ogbn-papers100m --solo-of 8`")
    args = ap.parse_args()

    if args.solo_of > 1:
        # sizing runs push toward the 288 GB HBM ceiling where caching-
        # allocator fragmentation (tens of GB reserved-but-unallocated)
        # turns a fitting run into an OOM. expandable_segments is not
        # supported by this ROCm build; cap large-block splitting and
        # garbage-collect the cache under pressure instead (both
        # supported by the HIP caching allocator)
        os.environ.setdefault(
            "PYTORCH_ALLOC_CONF",
            "max_split_size_mb:512,garbage_collection_threshold:0.8")

    env_world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    world = max(env_world, 1)
    if world == 1 and args.gpus > 1:
        raise SystemExit("bench.py with --gpus N>1 must be launched via "
                         "torch.distributed.run (one rank per GPU)")

    use_cuda = torch.cuda.is_available()
    device = args.device or ("cuda" if use_cuda else "cpu")
    backend = args.backend or ("nccl" if device.startswith("cuda")
                               else "gloo")
    if device.startswith("cuda"):
        if args.device is None:
            local_rank = int(os.environ.get("LOCAL_RANK", "0"))
            device = f"cuda:{local_rank}"
        torch.cuda.set_device(torch.device(device))

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend, rank=rank, world_size=world)

    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.graph.synthetic import SHAPES, synth_partition
    from pipegcn_amd.models.sage import GraphSAGE
    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.trainer import (exchange_halo_values, get_layer_size,
                                     precompute)
    from pipegcn_amd.utils.timer import comm_timer
    import torch.nn.functional as F
    import types

    import sys

    def phase(msg):
        if rank == 0:
            print(f"[bench t={time.time() - t_start:.0f}s] {msg}",
                  file=sys.stderr, flush=True)

    t_start = time.time()
    torch.manual_seed(0)
    # batch norm normalizes by the GLOBAL train count (reference semantics,
    # consistent only when every node is a train node — the inductive setup)
    train_frac = 1.0 if args.norm == "batch" else 0.66
    solo = args.solo_of > 1
    if solo and world > 1:
        raise SystemExit("--solo-of is a single-process sizing mode")
    part = synth_partition(args.shape, rank, args.solo_of if solo else world,
                           seed=0, train_frac=train_frac)
    phase(f"partition generated: {part.num_in} inner nodes, "
          f"{part.edges.shape[1]} edges, {part.halo_gnid.numel()} halo")
    rp = build_runtime_partition(part, device=device, solo=solo)
    phase("runtime partition built (CSR+CSC on device)")

    pipeline = not args.no_pipeline
    norm = None if args.norm == "none" else args.norm
    multilabel = SHAPES[args.shape][4]
    layer_size = get_layer_size(part.n_feat, args.n_hidden, part.n_class,
                                args.n_layers)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    comm_group = dist.new_group(backend=backend) if world > 1 else None
    ctx.buffer.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                           layer_size[: args.n_layers], use_pp=args.use_pp,
                           backend=backend, pipeline=pipeline,
                           corr_feat=args.feat_corr,
                           corr_grad=args.grad_corr,
                           device=device, group=comm_group,
                           collect_stats=True, dtype=dtype,
                           solo_world=args.solo_of)

    if args.model == "gcn":
        from pipegcn_amd.models.gcn import GCN

        if args.use_pp:
            raise SystemExit("--use-pp supports graphsage only "
                             "(reference parity)")
        model = GCN(layer_size, F.relu, use_pp=False, dropout=0.5,
                    norm=norm, n_linear=0,
                    train_size=part.n_train).to(device)
    else:
        model = GraphSAGE(layer_size, F.relu, use_pp=args.use_pp,
                          dropout=0.5, norm=norm, n_linear=0,
                          train_size=part.n_train).to(device)
    if dtype != torch.float32:
        model = model.to(dtype)
    ctx.reducer.init(model)
    if multilabel:
        loss_fcn = torch.nn.BCEWithLogitsLoss(reduction="sum")
    else:
        loss_fcn = torch.nn.CrossEntropyLoss(reduction="sum")
    optimizer = torch.optim.Adam(model.parameters(), lr=0.01)

    if dtype != torch.float32:
        rp.ndata["feat"] = rp.ndata["feat"].to(dtype)
    feat = rp.ndata["feat"]
    if args.use_pp:
        pp_args = types.SimpleNamespace(model="graphsage")
        feat = precompute(rp, pp_args)
        # bench never evals: the raw [num_all, F] feature tensor is dead
        # once the precomputed [num_in, 2F] input exists (papers100M
        # sizing: −14.1 GB of the 288 GB budget)
        rp.ndata.pop("feat", None)
        phase("use-pp precompute done")
    in_deg = rp.ndata["in_degree"]
    if args.model == "gcn":
        # GCN's symmetric normalization needs halo-node degrees too
        in_deg = exchange_halo_values(rp, in_deg)
    labels = rp.ndata["label"][: rp.num_train]
    model.train()

    def step():
        logits = model(rp.graph, feat, in_deg).float()
        loss = loss_fcn(logits[: rp.num_train], labels)
        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        ctx.buffer.next_epoch()
        ctx.reducer.synchronize(part.n_train)
        optimizer.step()
        comm_timer.clear()
        if solo and ctx.buffer._epoch % 4 == 0:
            # sizing mode runs within a few % of the memory ceiling:
            # periodically return freed large blocks to HIP so
            # fragmentation cannot accumulate (every epoch measured
            # 2.3x epoch-time from hipMalloc refill; every 4th amortizes
            # it; the timed headline configs never take this branch)
            torch.cuda.empty_cache()
        return loss

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    for w in range(args.warmup):
        step()
        phase(f"warmup {w + 1}/{args.warmup}" + (
            f" (peak {torch.cuda.max_memory_allocated() / 2**30:.1f} GB)"
            if device.startswith("cuda") else ""))
    if device.startswith("cuda"):
        torch.cuda.reset_peak_memory_stats()
    ctx.buffer.pop_comm_stats()  # reset overlap stats
    wait_s = 0.0
    barrier_sync()
    t0 = time.time()
    for _ in range(args.steps):
        logits = model(rp.graph, feat, in_deg).float()
        loss = loss_fcn(logits[: rp.num_train], labels)
        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        ctx.buffer.next_epoch()
        ctx.reducer.synchronize(part.n_train)
        optimizer.step()
        wait_s += comm_timer.tot_time()
        comm_timer.clear()
        if solo and ctx.buffer._epoch % 4 == 0:
            torch.cuda.empty_cache()  # see step(): sizing-mode only
    # drain the pipelined buffer queue so the final epoch's boundary
    # transfers are inside the timed region (they belong to the epoch)
    ctx.buffer.synchronize()
    barrier_sync()
    elapsed = time.time() - t0
    comm_busy = ctx.buffer.pop_comm_stats()

    # whole-job epoch time = max over ranks (NCCL needs device tensors)
    t = torch.tensor([elapsed, comm_busy, wait_s], dtype=torch.float64,
                     device=device if backend == "nccl" else "cpu")
    if world > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed, comm_busy, wait_s = t.cpu().tolist()
    epoch_s = elapsed / args.steps
    overlap_pct = (100.0 * max(comm_busy - wait_s, 0.0) / comm_busy
                   if comm_busy > 1e-9 else None)

    if rank == 0:
        n, avg_deg, n_feat, n_class, _ = SHAPES[args.shape]
        baseline = 0.2660  # BASELINE.md Reddit epoch time (rank 0), other hw
        result = {
            "metric": (f"epoch time (s), {args.shape} "
                       f"{args.n_layers}-layer "
                       f"{'GCN' if args.model == 'gcn' else 'GraphSAGE'}"
                       " (full-graph, "
                       f"{'pipelined' if pipeline else 'vanilla'})"),
            "value": epoch_s,
            "unit": "s/epoch",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": epoch_s * 1e3,
            "higher_is_better": False,
            "scaling": "strong",
            # only the reference-parity config (reddit shape, fp32) claims
            # a baseline ratio — reduced precision must not inflate it
            "vs_baseline": (epoch_s / baseline
                            if args.shape == "reddit"
                            and args.dtype == "fp32" and not solo else None),
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": f"{args.model}-{args.n_layers}L-h{args.n_hidden}",
                "graph": args.shape,
                "num_nodes": n,
                "avg_degree": avg_deg,
                "n_feat": n_feat,
                "n_class": n_class,
                "pipeline": pipeline,
                "use_pp": args.use_pp,
                "norm": args.norm,
                "parallelism": (f"solo rank0-of-{args.solo_of} (sizing)"
                                if solo else
                                f"graph-partition dp{world}"),
                "num_halo": rp.num_all - rp.num_in,
                "num_edges": rp.graph.csr.nnz,
                "comm_busy_s_per_epoch": comm_busy / args.steps,
                "comm_wait_s_per_epoch": wait_s / args.steps,
                "boundary_comm_overlap_pct": overlap_pct,
                "peak_mem_gb": (round(torch.cuda.max_memory_allocated()
                                      / 2**30, 2)
                                if device.startswith("cuda") else None),
            },
        }
        print(json.dumps(result))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
