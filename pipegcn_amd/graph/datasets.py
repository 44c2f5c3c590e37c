"""Dataset loading.

The reference pulls Reddit via DGL, ogbn-* via OGB and Yelp from npz/json
(/root/reference/helper/utils.py:17-96). Neither DGL nor OGB nor network
access exists in this environment, so this module supports:

 1. the reference's on-disk Yelp layout (adj_full.npz + feats.npy +
    class_map.json + role.json under dataset/yelp/), via scipy;
 2. a documented generic npz layout dataset/<name>.npz with arrays
    src, dst (int64 edges, no self-loops required), feat (float32 [N,F]),
    label (int64 [N] or float32 [N,C]), train/val/test_mask (bool [N]);
 3. synthetic graphs of the named shapes (reddit / ogbn-products / yelp /
    ogbn-papers100m and test sizes small/tiny) when no files are present —
    with a warning, since the benchmark contract is synthetic data of the
    named shapes.

Normalization matches the reference (/root/reference/helper/utils.py:93-95):
self-loops are removed and re-added exactly once.
"""
from __future__ import annotations

import json
import os
import warnings
from typing import Dict, Tuple

import torch

from pipegcn_amd.graph import synthetic

Graph = Tuple[torch.Tensor, torch.Tensor, int, Dict[str, torch.Tensor]]


def _normalize_self_loops(u, v, n):
    keep = u != v
    u, v = u[keep], v[keep]
    loop = torch.arange(n, dtype=torch.long)
    return torch.cat([u, loop]), torch.cat([v, loop])


def _load_npz(path: str) -> Graph:
    import numpy as np

    d = np.load(path)
    u = torch.from_numpy(d["src"]).long()
    v = torch.from_numpy(d["dst"]).long()
    feat = torch.from_numpy(d["feat"]).float()
    label = torch.from_numpy(d["label"])
    label = label.float() if label.ndim > 1 else label.long()
    n = feat.shape[0]
    ndata = {
        "feat": feat,
        "label": label,
        "train_mask": torch.from_numpy(d["train_mask"]).bool(),
        "val_mask": torch.from_numpy(d["val_mask"]).bool(),
        "test_mask": torch.from_numpy(d["test_mask"]).bool(),
    }
    u, v = _normalize_self_loops(u, v, n)
    return u, v, n, ndata


def _load_yelp(prefix: str = "dataset/yelp/") -> Graph:
    """Reference Yelp layout (/root/reference/helper/utils.py:33-71)."""
    import numpy as np
    import scipy.sparse as sp
    from sklearn.preprocessing import StandardScaler

    adj = sp.load_npz(os.path.join(prefix, "adj_full.npz"))
    n = adj.shape[0]
    coo = adj.tocoo()
    u = torch.from_numpy(coo.row).long()
    v = torch.from_numpy(coo.col).long()

    feats = np.load(os.path.join(prefix, "feats.npy"))
    with open(os.path.join(prefix, "class_map.json")) as f:
        class_map = json.load(f)
    with open(os.path.join(prefix, "role.json")) as f:
        role = json.load(f)

    labels = np.zeros((n, len(next(iter(class_map.values())))),
                      dtype=np.float32)
    for k, val in class_map.items():
        labels[int(k)] = val

    train_mask = torch.zeros(n, dtype=torch.bool)
    val_mask = torch.zeros(n, dtype=torch.bool)
    test_mask = torch.zeros(n, dtype=torch.bool)
    train_mask[role["tr"]] = True
    val_mask[role["va"]] = True
    test_mask[role["te"]] = True
    assert not (train_mask & val_mask).any()
    assert not (train_mask & test_mask).any()
    assert not (val_mask & test_mask).any()
    assert (train_mask | val_mask | test_mask).all()

    scaler = StandardScaler()
    scaler.fit(feats[train_mask.numpy()])
    feats = scaler.transform(feats)

    ndata = {
        "feat": torch.from_numpy(feats).float(),
        "label": torch.from_numpy(labels),
        "train_mask": train_mask,
        "val_mask": val_mask,
        "test_mask": test_mask,
    }
    u, v = _normalize_self_loops(u, v, n)
    return u, v, n, ndata


def load_data(dataset: str, nparts_hint: int = 4, seed: int = 0) -> Graph:
    """Return (u, v, num_nodes, ndata) with self-loops added."""
    name = dataset.lower()
    if name.startswith("synth-"):
        shape = name[len("synth-"):]
        u, v, n, ndata = synthetic.synth_global(
            shape, nparts_hint=nparts_hint, seed=seed)
        return u, v, n, ndata
    npz = os.path.join("dataset", f"{name}.npz")
    if os.path.exists(npz):
        return _load_npz(npz)
    if name == "yelp" and os.path.exists("dataset/yelp/adj_full.npz"):
        return _load_yelp()
    if name in synthetic.SHAPES:
        warnings.warn(
            f"dataset files for {dataset!r} not found under dataset/ — "
            f"generating a SYNTHETIC graph of the {dataset!r} shape "
            "(no network access in this environment).")
        return synthetic.synth_global(name, nparts_hint=nparts_hint,
                                      seed=seed)
    raise ValueError(f"Unknown dataset: {dataset}")


def data_stats(ndata) -> Tuple[int, int, int]:
    """(n_feat, n_class, n_train) from node data."""
    label = ndata["label"]
    n_feat = ndata["feat"].shape[1]
    n_class = (label.shape[1] if label.dim() > 1
               else int(label.max().item()) + 1)
    n_train = int(ndata["train_mask"].sum().item())
    return n_feat, n_class, n_train
