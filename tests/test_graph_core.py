"""Unit tests for the native graph core (CSR build, partitioner, CPU SpMM)."""
import torch

from pipegcn_amd import native
from pipegcn_amd.graph.csr import CSR, FullGraph, HaloGraph


def random_coo(n_src, n_dst, e, seed=0):
    g = torch.Generator().manual_seed(seed)
    u = torch.randint(0, n_src, (e,), generator=g)
    v = torch.randint(0, n_dst, (e,), generator=g)
    return u, v


def spmm_torch_ref(u, v, feat, num_rows, scale=None):
    out = torch.zeros(num_rows, feat.shape[1])
    out.index_add_(0, v, feat[u])
    if scale is not None:
        out *= scale.unsqueeze(1)
    return out


def test_build_csr_roundtrip():
    u, v = random_coo(50, 30, 400)
    indptr, indices = native().build_csr(u, v, 30)
    assert indptr[-1].item() == 400
    # every edge present exactly once
    got = sorted((indices[e].item(), r) for r in range(30)
                 for e in range(indptr[r], indptr[r + 1]))
    want = sorted(zip(u.tolist(), v.tolist()))
    assert got == want


def test_spmm_cpu_matches_torch():
    for f in (1, 3, 8, 33, 64):
        u, v = random_coo(40, 25, 300, seed=f)
        feat = torch.randn(40, f)
        scale = torch.rand(25) + 0.5
        csr = CSR.from_coo(u, v, 25, 40)
        out = native().spmm(csr.indptr, csr.indices, feat, scale, torch.Tensor(), torch.Tensor(), 25)
        ref = spmm_torch_ref(u, v, feat, 25, scale)
        assert torch.allclose(out, ref, atol=1e-5), f"F={f}"


def test_spmm_no_scale():
    u, v = random_coo(40, 25, 300)
    feat = torch.randn(40, 8)
    csr = CSR.from_coo(u, v, 25, 40)
    out = native().spmm(csr.indptr, csr.indices, feat, torch.Tensor(), torch.Tensor(), torch.Tensor(), 25)
    assert torch.allclose(out, spmm_torch_ref(u, v, feat, 25), atol=1e-5)


def test_partitioner_balance_and_cut():
    # community graph: partitioner should find communities
    n, k = 400, 4
    g = torch.Generator().manual_seed(3)
    dst = torch.randint(0, n, (6000,), generator=g)
    comm = dst // (n // k)
    local = torch.rand(6000, generator=g) < 0.9
    src = torch.where(
        local,
        comm * (n // k) + torch.randint(0, n // k, (6000,), generator=g),
        torch.randint(0, n, (6000,), generator=g))
    su, sv = torch.cat([src, dst]), torch.cat([dst, src])
    indptr, indices = native().build_csr(su, sv, n)
    for obj in (0, 1):
        part = native().partition_graph(indptr, indices, k, obj, 0.05, 8, 0)
        sizes = torch.bincount(part.long(), minlength=k)
        assert sizes.min() >= n / k * 0.9, sizes
        assert sizes.max() <= n / k * 1.1, sizes
        cut = (part[src.int().long()] != part[dst.long()]).float().mean()
        assert cut < 0.5, f"cut fraction {cut} too high (random ~0.75)"


def test_partitioner_single_part():
    u, v = random_coo(20, 20, 50)
    indptr, indices = native().build_csr(u, v, 20)
    part = native().partition_graph(indptr, indices, 1, 0, 0.05, 4, 0)
    assert (part == 0).all()


def test_halo_graph_csc_is_transpose():
    u, v = random_coo(35, 20, 200)
    g = HaloGraph.from_edges(u, v, 20, 35)
    feat = torch.randn(35, 5)
    gout = torch.randn(20, 5)
    # <A x, g> == <x, A^T g>
    ax = native().spmm(g.csr.indptr, g.csr.indices, feat, torch.Tensor(),
                       torch.Tensor(), torch.Tensor(), 20)
    atg = native().spmm(g.csc.indptr, g.csc.indices, gout, torch.Tensor(),
                        torch.Tensor(), torch.Tensor(), 35)
    assert torch.allclose((ax * gout).sum(), (feat * atg).sum(), atol=1e-3)


def test_full_graph_degrees():
    u, v = random_coo(30, 30, 100)
    fg = FullGraph.from_coo(u, v, 30)
    ref = torch.bincount(v, minlength=30).float()
    assert torch.equal(fg.in_degrees(), ref)


def test_partitioner_deterministic():
    u, v = random_coo(500, 500, 4000, seed=9)
    su, sv = torch.cat([u, v]), torch.cat([v, u])
    indptr, indices = native().build_csr(su, sv, 500)
    a = native().partition_graph(indptr, indices, 4, 1, 0.05, 8, 7)
    b = native().partition_graph(indptr, indices, 4, 1, 0.05, 8, 7)
    assert torch.equal(a, b), "same seed must give the same partition"
    c = native().partition_graph(indptr, indices, 4, 1, 0.05, 8, 8)
    # different seed may legitimately differ (not asserted equal)
    assert c.shape == a.shape


def test_csr_row_order_is_lpt():
    u, v = random_coo(60, 40, 500, seed=4)
    csr = CSR.from_coo(u, v, 40, 60)
    deg = (csr.indptr[1:] - csr.indptr[:-1])
    ordered = deg[csr.row_order.long()]
    assert (ordered[:-1] >= ordered[1:]).all(), \
        "row_order must sort degrees descending"
    assert sorted(csr.row_order.tolist()) == list(range(40))


def test_synth_partition_invariants():
    from pipegcn_amd.graph.synthetic import synth_partition

    parts = [synth_partition("small", r, 3, seed=2) for r in range(3)]
    assert sum(p.num_in for p in parts) == 1000
    for p in parts:
        assert int(p.ndata["in_degree"].sum()) == p.edges.shape[1]
        assert p.edges[1].max() < p.num_in  # dsts are inner
        assert p.edges[0].max() < p.num_local


def _comm_volume(u, v, n, part):
    """Total communication volume: sum over nodes of the number of OTHER
    partitions holding at least one neighbor (= halo replicas the runtime
    will exchange; METIS objtype='vol' semantics,
    /root/reference/helper/utils.py:143)."""
    su, sv = torch.cat([u, v]), torch.cat([v, u])
    keep = su != sv
    su, sv = su[keep], sv[keep]
    pairs = torch.unique(torch.stack([su, part[sv].long()]), dim=1)
    return int((pairs[1] != part[pairs[0]].long()).sum())


def test_exact_vol_gain_verified_and_better():
    """PIPEGCN_PART_CHECK_VOL=1 makes the C++ refinement TORCH_CHECK every
    accepted move's exact volume gain against a brute-force local
    recomputation; the exact objective must also beat the approximate
    (own-replica-delta-only) gain on communication volume."""
    import os

    from pipegcn_amd.graph.partition import assign_partitions

    torch.manual_seed(0)
    n, e = 3000, 30000
    u = torch.randint(0, n, (e,))
    v = torch.randint(0, n, (e,))
    try:
        for seed in (0, 1):
            os.environ["PIPEGCN_PART_EXACT_VOL_MB"] = "0"  # force approx
            pa = assign_partitions(u, v, n, 4, "metis", "vol", seed)
            os.environ["PIPEGCN_PART_EXACT_VOL_MB"] = "4096"
            os.environ["PIPEGCN_PART_CHECK_VOL"] = "1"  # oracle on
            pe = assign_partitions(u, v, n, 4, "metis", "vol", seed)
            va = _comm_volume(u, v, n, pa)
            ve = _comm_volume(u, v, n, pe)
            assert ve < va, (seed, ve, va)
            # balance (0.05 slack + single-node rounding)
            sizes = torch.bincount(pe.long(), minlength=4)
            assert sizes.max() <= int(n / 4 * 1.05) + 1
            assert sizes.min() >= int(n / 4 * 0.94)
    finally:
        os.environ.pop("PIPEGCN_PART_CHECK_VOL", None)
        os.environ.pop("PIPEGCN_PART_EXACT_VOL_MB", None)


def test_partitioner_thread_count_invariance():
    """The parallel-prefilter refinement and parallel coarsening must give
    the SAME partition regardless of thread count (evaluation is a pure
    function of the round snapshot; application is serial in node order)."""
    import subprocess
    import sys

    code = """
import torch
torch.manual_seed(0)
torch.set_num_threads({n})
from pipegcn_amd.graph.partition import assign_partitions
u = torch.randint(0, 4000, (30000,), generator=torch.Generator().manual_seed(1))
v = torch.randint(0, 4000, (30000,), generator=torch.Generator().manual_seed(2))
p = assign_partitions(u, v, 4000, 4, "metis", "vol", 3)
print(hash(tuple(p.tolist())))
"""
    outs = []
    for n in (1, 8):
        r = subprocess.run([sys.executable, "-c", code.format(n=n)],
                           capture_output=True, text=True)
        assert r.returncode == 0, r.stderr
        outs.append(r.stdout.strip())
    assert outs[0] == outs[1], outs


def test_suitor_matching_deterministic_and_sane():
    """Force the parallel Suitor matching path (PIPEGCN_PART_SUITOR_MIN)
    and check thread-count determinism + partition sanity on it."""
    import subprocess
    import sys

    code = """
import os, torch
os.environ["PIPEGCN_PART_SUITOR_MIN"] = "1000"  # force Suitor
torch.set_num_threads({n})
from pipegcn_amd.graph.partition import assign_partitions
g1 = torch.Generator().manual_seed(11)
u = torch.randint(0, 20000, (160000,), generator=g1)
v = torch.randint(0, 20000, (160000,), generator=g1)
p = assign_partitions(u, v, 20000, 4, "metis", "vol", 3)
sizes = torch.bincount(p.long(), minlength=4)
assert sizes.max() <= int(20000 / 4 * 1.05) + 1, sizes
print(hash(tuple(p.tolist())))
"""
    outs = []
    for n in (1, 8):
        r = subprocess.run([sys.executable, "-c", code.format(n=n)],
                           capture_output=True, text=True)
        assert r.returncode == 0, r.stderr
        outs.append(r.stdout.strip())
    assert outs[0] == outs[1], outs
