"""Launcher (reference: /root/reference/main.py).

Parses args, names the partition directory, runs partitioning once (node
rank 0), then spawns one process per local partition. Unlike the reference,
the 'nccl' backend (RCCL over xGMI) is fully supported: each process is
pinned to one GPU via CUDA_VISIBLE_DEVICES round-robin, exactly like the
gloo path.
"""
import os
import random
import warnings

import torch
import torch.multiprocessing as mp

from pipegcn_amd import trainer
from pipegcn_amd.cli import create_parser
from pipegcn_amd.graph import datasets
from pipegcn_amd.graph.partition import partition_and_save

if __name__ == "__main__":
    args = create_parser()
    if args.fix_seed is False:
        if args.parts_per_node < args.n_partitions:
            warnings.warn("multi-node runs need identical seeds on every "
                          "node — pass --fix-seed")
        args.seed = random.randint(0, 1 << 31)

    if args.graph_name == "":
        mode = "induc" if args.inductive else "trans"
        args.graph_name = "%s-%d-%s-%s-%s" % (
            args.dataset, args.n_partitions, args.partition_method,
            args.partition_obj, mode)
    graph_dir = os.path.join("partitions", args.graph_name)

    if args.skip_partition:
        if args.n_feat == 0 or args.n_class == 0 or args.n_train == 0:
            warnings.warn("--skip-partition without --n-feat/--n-class/"
                          "--n-train forces a full data load just to count "
                          "them — pass all three to skip it")
            u, v, n, ndata = datasets.load_data(
                args.dataset, nparts_hint=args.n_partitions, seed=args.seed)
            args.n_feat, args.n_class, args.n_train = datasets.data_stats(
                ndata)
    else:
        u, v, n, ndata = datasets.load_data(
            args.dataset, nparts_hint=args.n_partitions, seed=args.seed)
        args.n_feat, args.n_class, args.n_train = datasets.data_stats(ndata)
        if args.node_rank == 0:
            if args.inductive:
                su, sv, sn, snd = trainer.node_subgraph(
                    u, v, n, ndata, ndata["train_mask"])
                partition_and_save(su, sv, sn, snd, graph_dir,
                                   args.n_partitions, args.partition_method,
                                   args.partition_obj, args.seed)
            else:
                partition_and_save(u, v, n, ndata, graph_dir,
                                   args.n_partitions, args.partition_method,
                                   args.partition_obj, args.seed)
        del u, v, ndata

    print(args)

    if args.backend == "nccl" and not torch.cuda.is_available():
        raise SystemExit("--backend nccl (RCCL) needs GPUs; "
                         "use --backend gloo on CPU")

    if args.backend in ("gloo", "nccl"):
        if "CUDA_VISIBLE_DEVICES" in os.environ:
            devices = os.environ["CUDA_VISIBLE_DEVICES"].split(",")
        elif torch.cuda.is_available():
            devices = [f"{i}" for i in range(torch.cuda.device_count())]
        else:
            devices = []
        mp.set_start_method("spawn", force=True)
        processes = []
        start_id = args.node_rank * args.parts_per_node
        for i in range(start_id,
                       min(start_id + args.parts_per_node,
                           args.n_partitions)):
            if devices:
                os.environ["CUDA_VISIBLE_DEVICES"] = devices[
                    i % len(devices)]
            p = mp.Process(target=trainer.init_processes,
                           args=(i, args.n_partitions, args))
            p.start()
            processes.append(p)
        for p in processes:
            p.join()
    elif args.backend == "mpi":
        raise NotImplementedError("mpi backend is not supported; use "
                                  "'nccl' (RCCL) or 'gloo'")
    else:
        raise ValueError(f"unknown backend {args.backend}")
