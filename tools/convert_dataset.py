#!/usr/bin/env python3
"""Convert public dataset releases into pipegcn_amd's generic npz layout.

The reference ingests real data through DGL/OGB package loaders
(/root/reference/helper/utils.py:17-96: RedditDataset,
DglNodePropPredDataset, the GraphSAINT Yelp layout). Those packages (and
network access) are not part of this framework's runtime, so this tool
parses the *published on-disk release formats directly* and writes the
documented generic layout that `pipegcn_amd.graph.datasets._load_npz`
reads:

    dataset/<name>.npz with arrays
        src, dst     int64 [E]      directed edge list (no self loops req.)
        feat         float32 [N,F]
        label        int64 [N] (single-label) or float32 [N,C] (multilabel)
        train_mask / val_mask / test_mask   bool [N]

Self-loop normalization (remove + re-add exactly once,
/root/reference/helper/utils.py:93-95) happens at LOAD time, not here.

Supported source layouts (what the public downloads actually unpack to):

  reddit      DGL release: <src>/reddit_data.npz (feature, label,
              node_types: 1=train 2=val 3=test) + <src>/reddit_graph.npz
              (scipy.sparse symmetric adjacency).
  ogb-csv     OGB csv release (ogbn-products, ogbn-arxiv): <src>/raw/
              {edge.csv.gz, node-feat.csv.gz, node-label.csv.gz,
              num-node-list.csv.gz} + <src>/split/<scheme>/
              {train,valid,test}.csv.gz. Pass --add-inverse-edges for
              undirected releases (ogbn-products: OGB's own loader adds
              them; master.csv add_inverse_edge=True).
  ogb-binary  OGB binary release (ogbn-papers100M): <src>/raw/data.npz
              (edge_index int [2,E], node_feat float [N,F]) +
              <src>/raw/node-label.npz (node_label; NaN = unlabeled) +
              the same split/<scheme>/*.csv.gz. Needs ~150 GB host RAM
              for papers100M (57 GB feat + 26 GB edges, plus the write).
  yelp        GraphSAINT layout (adj_full.npz + feats.npy +
              class_map.json + role.json) — normally loaded natively by
              datasets._load_yelp; conversion here is optional and skips
              the train-fitted StandardScaler (the native loader applies
              it at load time; the npz layout stores raw features, so
              prefer the native path for Yelp).

Usage:
    python tools/convert_dataset.py reddit     --src dl/reddit    --out dataset/reddit.npz
    python tools/convert_dataset.py ogb-csv    --src dl/products  --out dataset/ogbn-products.npz --add-inverse-edges
    python tools/convert_dataset.py ogb-binary --src dl/papers    --out dataset/ogbn-papers100m.npz
"""
from __future__ import annotations

import argparse
import glob
import gzip
import json
import os

import numpy as np


def _read_int_csv_gz(path: str) -> np.ndarray:
    """Read a (possibly multi-column) integer csv.gz with no header."""
    with gzip.open(path, "rt") as f:
        return np.loadtxt(f, dtype=np.int64, delimiter=",", ndmin=2)


def _read_float_csv_gz(path: str) -> np.ndarray:
    with gzip.open(path, "rt") as f:
        return np.loadtxt(f, dtype=np.float32, delimiter=",", ndmin=2)


def _find_split_dir(src: str) -> str:
    """OGB releases have exactly one split scheme dir (sales_ranking/time)."""
    cands = sorted(d for d in glob.glob(os.path.join(src, "split", "*"))
                   if os.path.isdir(d))
    if len(cands) != 1:
        raise FileNotFoundError(
            f"expected exactly one split scheme under {src}/split, "
            f"found {cands}")
    return cands[0]


def _split_masks(split_dir: str, n: int):
    masks = {}
    for part, fname in (("train_mask", "train.csv.gz"),
                        ("val_mask", "valid.csv.gz"),
                        ("test_mask", "test.csv.gz")):
        idx = _read_int_csv_gz(os.path.join(split_dir, fname)).ravel()
        m = np.zeros(n, dtype=bool)
        m[idx] = True
        masks[part] = m
    return masks


def convert_reddit(src: str):
    import scipy.sparse as sp

    data = np.load(os.path.join(src, "reddit_data.npz"))
    adj = sp.load_npz(os.path.join(src, "reddit_graph.npz")).tocoo()
    feat = data["feature"].astype(np.float32)
    label = data["label"].astype(np.int64)
    node_types = data["node_types"]
    n = feat.shape[0]
    assert adj.shape == (n, n), (adj.shape, n)
    return dict(src=adj.row.astype(np.int64), dst=adj.col.astype(np.int64),
                feat=feat, label=label,
                train_mask=node_types == 1,
                val_mask=node_types == 2,
                test_mask=node_types == 3)


def _finish_ogb(edges, feat, label, src_dir, add_inverse):
    n = feat.shape[0]
    u, v = edges[0].astype(np.int64), edges[1].astype(np.int64)
    if add_inverse:
        u, v = np.concatenate([u, v]), np.concatenate([v, u])
    # single-label node classification: labels come as float with NaN on
    # unlabeled nodes (papers100M); masks never select those
    if label.ndim == 2 and label.shape[1] == 1:
        label = label.ravel()
    label = np.nan_to_num(label, nan=-1.0).astype(np.int64)
    out = dict(src=u, dst=v, feat=feat.astype(np.float32, copy=False),
               label=label)
    out.update(_split_masks(_find_split_dir(src_dir), n))
    return out


def convert_ogb_csv(src: str, add_inverse: bool):
    raw = os.path.join(src, "raw")
    n = int(_read_int_csv_gz(os.path.join(raw, "num-node-list.csv.gz"))[0, 0])
    edge = _read_int_csv_gz(os.path.join(raw, "edge.csv.gz")).T
    feat = _read_float_csv_gz(os.path.join(raw, "node-feat.csv.gz"))
    label = _read_float_csv_gz(os.path.join(raw, "node-label.csv.gz"))
    assert feat.shape[0] == n, (feat.shape, n)
    return _finish_ogb(edge, feat, label, src, add_inverse)


def convert_ogb_binary(src: str, add_inverse: bool):
    raw = os.path.join(src, "raw")
    d = np.load(os.path.join(raw, "data.npz"))
    lab = np.load(os.path.join(raw, "node-label.npz"))
    return _finish_ogb(d["edge_index"], d["node_feat"],
                       lab["node_label"], src, add_inverse)


def convert_yelp(src: str):
    import scipy.sparse as sp

    adj = sp.load_npz(os.path.join(src, "adj_full.npz")).tocoo()
    feat = np.load(os.path.join(src, "feats.npy")).astype(np.float32)
    with open(os.path.join(src, "class_map.json")) as f:
        class_map = json.load(f)
    with open(os.path.join(src, "role.json")) as f:
        role = json.load(f)
    n = feat.shape[0]
    c = len(next(iter(class_map.values())))
    label = np.zeros((n, c), dtype=np.float32)
    for k, val in class_map.items():
        label[int(k)] = val
    masks = {m: np.zeros(n, dtype=bool)
             for m in ("train_mask", "val_mask", "test_mask")}
    masks["train_mask"][role["tr"]] = True
    masks["val_mask"][role["va"]] = True
    masks["test_mask"][role["te"]] = True
    return dict(src=adj.row.astype(np.int64), dst=adj.col.astype(np.int64),
                feat=feat, label=label, **masks)


def main():
    ap = argparse.ArgumentParser(description=__doc__.split("\n")[0])
    ap.add_argument("layout",
                    choices=["reddit", "ogb-csv", "ogb-binary", "yelp"])
    ap.add_argument("--src", required=True,
                    help="unpacked release directory")
    ap.add_argument("--out", required=True,
                    help="output path, e.g. dataset/reddit.npz")
    ap.add_argument("--add-inverse-edges", action="store_true",
                    help="symmetrize the edge list (undirected OGB "
                         "releases, e.g. ogbn-products)")
    ap.add_argument("--compress", action="store_true",
                    help="np.savez_compressed (slower, smaller)")
    args = ap.parse_args()

    if args.layout == "reddit":
        arrays = convert_reddit(args.src)
    elif args.layout == "ogb-csv":
        arrays = convert_ogb_csv(args.src, args.add_inverse_edges)
    elif args.layout == "ogb-binary":
        arrays = convert_ogb_binary(args.src, args.add_inverse_edges)
    else:
        arrays = convert_yelp(args.src)

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    save = np.savez_compressed if args.compress else np.savez
    save(args.out, **arrays)
    n, f = arrays["feat"].shape
    lab = arrays["label"]
    n_class = lab.shape[1] if lab.ndim > 1 else int(lab.max()) + 1
    print(f"wrote {args.out}: {n} nodes, {arrays['src'].size} directed "
          f"edges, {f} features, {n_class} classes, "
          f"{int(arrays['train_mask'].sum())} train")


if __name__ == "__main__":
    main()
