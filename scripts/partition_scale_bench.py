#!/usr/bin/env python3
"""papers100M-scale partitioner benchmark (VERDICT round-1 item 7).

Generates a 111M-node community-structured synthetic graph directly in
symmetrized-COO form (chunk-friendly, int64), builds the CSR, runs the
native multilevel partitioner with the exact communication-volume
objective, and reports wall time + quality (volume / cut / balance)
against the planted-community assignment.

The edge budget is configurable: the full papers100M shape (avg directed
degree 29 -> 6.4B symmetrized edges) needs ~150 GB host RAM; the default
--avg-deg 8 (~1.8B symmetrized edges) exercises the same 111M-node scale
within a 64 GB box. Node count is the scale driver for every serial
phase (matching order, refinement sweeps, counters).

Usage: PIPEGCN_PART_VERBOSE=1 python scripts/partition_scale_bench.py \
           [--nodes 111059956] [--avg-deg 8] [--nparts 8]
"""
import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from pipegcn_amd import native  # noqa: E402


def log(msg):
    print(f"[scale t={time.time() - T0:6.0f}s] {msg}", file=sys.stderr,
          flush=True)


def gen_coo(n, avg_deg, nparts_hint, frac_remote, seed):
    """Locality-structured COO: sources cluster around their destination
    with power-law distances (real citation/social graphs — papers100M
    included — have strong id-locality after their release orderings;
    uniform-random 'communities' would instead be the pathological
    no-local-structure worst case for ANY multilevel partitioner,
    METIS included). frac_remote of edges are global-uniform."""
    g = torch.Generator().manual_seed(seed)
    E = n * avg_deg
    dst = torch.repeat_interleave(torch.arange(n), avg_deg)
    # power-law offset magnitude: P(d) ~ d^-1.5 up to n
    mag = (torch.rand(E, generator=g).clamp(min=1e-9) ** -2.0).long() + 1
    sign = torch.where(torch.rand(E, generator=g) < 0.5, -1, 1)
    src = (dst + sign * mag).remainder(n)
    del mag, sign
    remote = torch.rand(E, generator=g) < frac_remote
    nr = int(remote.sum())
    src[remote] = (torch.rand(nr, generator=g) * n).long()
    del remote
    return src, dst


def metrics(indptr, indices, part, nparts, chunk=50_000_000):
    """Volume / cut / balance from the CSR, chunked over edges; the
    distinct (node, neighbor-part) pairs are tracked in an n*nparts
    bitset-style bool tensor (111 MB*8 at papers scale)."""
    n = indptr.numel() - 1
    seen = torch.zeros(n * nparts, dtype=torch.bool)
    cut = 0
    deg = indptr[1:] - indptr[:-1]
    dst_all = torch.repeat_interleave(torch.arange(n), deg)
    E = indices.numel()
    for s in range(0, E, chunk):
        e = min(s + chunk, E)
        u = dst_all[s:e]          # row (dst) node
        v = indices[s:e].long()   # neighbor
        keep = u != v
        u, v = u[keep], v[keep]
        pu, pv = part[u].long(), part[v].long()
        cut += int((pu != pv).sum()) // 2  # symmetric: each edge twice
        seen[u * nparts + pv] = True
    own = part.long().unsqueeze(1).expand(n, nparts)
    vol = int((seen.view(n, nparts)
               & (own != torch.arange(nparts))).sum())
    sizes = torch.bincount(part.long(), minlength=nparts)
    return vol, cut, sizes.tolist()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=111_059_956)
    ap.add_argument("--avg-deg", type=int, default=8)
    ap.add_argument("--nparts", type=int, default=8)
    ap.add_argument("--frac-remote", type=float, default=0.1)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--skip-metrics", action="store_true")
    args = ap.parse_args()
    n = args.nodes

    log(f"generating {n} nodes x deg {args.avg_deg}")
    src, dst = gen_coo(n, args.avg_deg, args.nparts, args.frac_remote,
                       args.seed)
    log(f"{src.numel()} directed edges; symmetrizing")
    su = torch.cat([src, dst])
    sv = torch.cat([dst, src])
    del src, dst
    log("building CSR")
    indptr, indices = native().build_csr(su, sv, n)
    del su, sv
    log(f"CSR done ({indices.numel()} directed edges); partitioning")

    t0 = time.time()
    part = native().partition_graph(indptr, indices, args.nparts, 1, 0.05,
                                    8, args.seed)
    t_part = time.time() - t0
    log(f"partition done in {t_part:.1f}s")

    result = {"nodes": n, "sym_edges": int(indices.numel()),
              "nparts": args.nparts, "partition_s": round(t_part, 1)}
    if not args.skip_metrics:
        vol, cut, sizes = metrics(indptr, indices, part, args.nparts)
        planted = ((torch.arange(n) * args.nparts) // n).to(torch.int32)
        pvol, pcut, psizes = metrics(indptr, indices, planted, args.nparts)
        result.update(volume=vol, cut=cut, sizes=sizes,
                      planted_volume=pvol, planted_cut=pcut,
                      vol_vs_planted=round(vol / max(pvol, 1), 3))
    print(json.dumps(result))


if __name__ == "__main__":
    T0 = time.time()
    main()
