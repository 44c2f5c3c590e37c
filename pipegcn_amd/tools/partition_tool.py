"""Standalone partitioner CLI (the partitioner-tool build item).

Produces the on-disk partition directory without training:

    python -m pipegcn_amd.tools.partition_tool --dataset reddit \
        --n-partitions 8 --partition-obj vol [--inductive]

The result is what main.py/--skip-partition and the trainer consume.
"""
import argparse
import os

from pipegcn_amd.graph import datasets
from pipegcn_amd.graph.partition import partition_and_save


def main():
    ap = argparse.ArgumentParser(description="pipegcn_amd graph partitioner")
    ap.add_argument("--dataset", type=str, required=True)
    ap.add_argument("--n-partitions", type=int, required=True)
    ap.add_argument("--partition-method", choices=["metis", "random"],
                    default="metis")
    ap.add_argument("--partition-obj", choices=["vol", "cut"], default="vol")
    ap.add_argument("--inductive", action="store_true")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--graph-name", type=str, default="")
    args = ap.parse_args()

    if not args.graph_name:
        mode = "induc" if args.inductive else "trans"
        args.graph_name = "%s-%d-%s-%s-%s" % (
            args.dataset, args.n_partitions, args.partition_method,
            args.partition_obj, mode)
    graph_dir = os.path.join("partitions", args.graph_name)

    u, v, n, ndata = datasets.load_data(args.dataset,
                                        nparts_hint=args.n_partitions,
                                        seed=args.seed)
    if args.inductive:
        from pipegcn_amd.trainer import node_subgraph

        u, v, n, ndata = node_subgraph(u, v, n, ndata, ndata["train_mask"])
    partition_and_save(u, v, n, ndata, graph_dir, args.n_partitions,
                       args.partition_method, args.partition_obj, args.seed)
    n_feat, n_class, n_train = datasets.data_stats(ndata)
    print(f"partitioned {args.dataset} ({n} nodes, {u.numel()} edges) into "
          f"{args.n_partitions} parts at {graph_dir} "
          f"(n_feat={n_feat} n_class={n_class} n_train={n_train})")


if __name__ == "__main__":
    main()
