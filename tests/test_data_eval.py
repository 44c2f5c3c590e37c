"""Dataset loading, evaluation-path, and partition-format robustness."""
import json
import os

import numpy as np
import pytest
import torch

from pipegcn_amd.graph import datasets, partition, synthetic
from pipegcn_amd.trainer import calc_acc, inductive_split, node_subgraph


def test_npz_dataset_roundtrip(tmp_path, monkeypatch):
    n, f, c = 50, 8, 3
    g = torch.Generator().manual_seed(0)
    src = torch.randint(0, n, (200,), generator=g)
    dst = torch.randint(0, n, (200,), generator=g)
    feat = torch.randn(n, f, generator=g)
    label = torch.randint(0, c, (n,), generator=g)
    r = torch.rand(n, generator=g)
    os.makedirs(tmp_path / "dataset")
    np.savez(tmp_path / "dataset" / "mygraph.npz",
             src=src.numpy(), dst=dst.numpy(), feat=feat.numpy().astype("f4"),
             label=label.numpy(),
             train_mask=(r < 0.5).numpy(), val_mask=((r >= 0.5) &
                                                     (r < 0.75)).numpy(),
             test_mask=(r >= 0.75).numpy())
    monkeypatch.chdir(tmp_path)
    u, v, nn_, ndata = datasets.load_data("mygraph")
    assert nn_ == n and ndata["feat"].shape == (n, f)
    # self-loops normalized: exactly one per node
    loops = (u == v).sum().item()
    assert loops == n
    n_feat, n_class, n_train = datasets.data_stats(ndata)
    assert (n_feat, n_class) == (f, c)


def test_synthetic_fallback_warns():
    with pytest.warns(UserWarning, match="SYNTHETIC"):
        u, v, n, ndata = datasets.load_data("reddit")
    assert n == 232_965


def test_calc_acc_multilabel():
    logits = torch.tensor([[2.0, -1.0], [-1.0, 3.0]])
    labels = torch.tensor([[1.0, 0.0], [0.0, 1.0]])
    assert calc_acc(logits, labels) == 1.0
    assert calc_acc(torch.tensor([[1.0, -2.0]]),
                    torch.tensor([0])) == 1.0


def test_node_subgraph_and_inductive_split():
    u, v, n, ndata = synthetic.synth_global("tiny", seed=3)
    mask = ndata["train_mask"]
    su, sv, sn, snd = node_subgraph(u, v, n, ndata, mask)
    assert sn == int(mask.sum())
    assert su.max() < sn and sv.max() < sn
    tr, va, full = inductive_split(u, v, n, ndata)
    assert tr[2] <= va[2] <= full[2]


def test_partition_idempotent(tmp_path):
    u, v, n, ndata = synthetic.synth_global("tiny", seed=1)
    d = str(tmp_path / "p")
    partition.partition_and_save(u, v, n, ndata, d, 2, "metis", "cut", 0)
    stamp = os.path.getmtime(os.path.join(d, "part0.pt"))
    partition.partition_and_save(u, v, n, ndata, d, 2, "metis", "cut", 0)
    assert os.path.getmtime(os.path.join(d, "part0.pt")) == stamp


def test_partition_covers_all_edges(tmp_path):
    u, v, n, ndata = synthetic.synth_global("small", seed=2)
    d = str(tmp_path / "p")
    partition.partition_and_save(u, v, n, ndata, d, 3, "metis", "vol", 0)
    with open(os.path.join(d, "meta.json")) as f:
        meta = json.load(f)
    tot_edges = 0
    tot_inner = 0
    for r in range(3):
        part = partition.load_partition(d, r)
        tot_edges += part.edges.shape[1]
        tot_inner += part.num_in
        # in-degree consistency: stored full-graph degree == local edge count
        deg = torch.bincount(part.edges[1], minlength=part.num_in).float()
        assert torch.equal(deg, part.ndata["in_degree"])
    assert tot_edges == meta["num_edges"]
    assert tot_inner == n


def test_eval_matches_train_forward():
    """Single partition, dropout 0: eval-path (FullGraph) logits must match
    the training-path (HaloGraph) logits on the same weights."""
    import torch.nn.functional as F

    from pipegcn_amd.graph.csr import FullGraph, HaloGraph
    from pipegcn_amd.models.sage import GraphSAGE

    from pipegcn_amd.parallel import context as ctx
    from pipegcn_amd.parallel.buffer import Buffer

    u, v, n, ndata = synthetic.synth_global("tiny", seed=7)
    hg = HaloGraph.from_edges(u, v, n, n)
    fg = FullGraph.from_coo(u, v, n, ndata)
    deg = torch.bincount(v, minlength=n).float()
    ctx.buffer = Buffer()
    ctx.buffer.init_buffer(n, n, [None], [None], [16, 8])
    torch.manual_seed(0)
    model = GraphSAGE([16, 8, 4], F.relu, use_pp=False, dropout=0.0,
                      norm="layer", n_linear=0)
    model.train()
    logits_tr = model(hg, ndata["feat"], deg)
    model.eval()
    with torch.no_grad():
        logits_ev = model(fg, ndata["feat"])
    assert torch.allclose(logits_tr, logits_ev, atol=1e-5), \
        (logits_tr - logits_ev).abs().max()


def test_yelp_reference_layout(tmp_path, monkeypatch):
    """Load the reference's on-disk Yelp format (adj_full.npz + feats.npy +
    class_map.json + role.json — /root/reference/helper/utils.py:33-71)."""
    import scipy.sparse as sp

    n, f, c = 40, 6, 4
    rng = np.random.RandomState(0)
    adj = sp.random(n, n, density=0.2, format="csr", random_state=rng)
    d = tmp_path / "dataset" / "yelp"
    os.makedirs(d)
    sp.save_npz(d / "adj_full.npz", adj)
    np.save(d / "feats.npy", rng.randn(n, f))
    cm = {str(i): rng.randint(0, 2, c).tolist() for i in range(n)}
    with open(d / "class_map.json", "w") as fh:
        json.dump(cm, fh)
    idx = rng.permutation(n)
    role = {"tr": idx[:20].tolist(), "va": idx[20:30].tolist(),
            "te": idx[30:].tolist()}
    with open(d / "role.json", "w") as fh:
        json.dump(role, fh)
    monkeypatch.chdir(tmp_path)
    u, v, nn_, ndata = datasets.load_data("yelp")
    assert nn_ == n
    assert ndata["label"].shape == (n, c)
    assert ndata["train_mask"].sum() == 20
    # features are StandardScaler-normalized over train rows
    tr = ndata["feat"][ndata["train_mask"]]
    assert tr.mean(0).abs().max() < 0.5


def test_npz_multilabel(tmp_path, monkeypatch):
    n, f, c = 30, 4, 3
    rng = np.random.RandomState(1)
    os.makedirs(tmp_path / "dataset")
    np.savez(tmp_path / "dataset" / "ml.npz",
             src=rng.randint(0, n, 80), dst=rng.randint(0, n, 80),
             feat=rng.randn(n, f).astype("f4"),
             label=rng.randint(0, 2, (n, c)).astype("f4"),
             train_mask=np.ones(n, bool), val_mask=np.zeros(n, bool),
             test_mask=np.zeros(n, bool))
    monkeypatch.chdir(tmp_path)
    u, v, nn_, ndata = datasets.load_data("ml")
    assert ndata["label"].dtype == torch.float32
    n_feat, n_class, n_train = datasets.data_stats(ndata)
    assert n_class == c and n_train == n
