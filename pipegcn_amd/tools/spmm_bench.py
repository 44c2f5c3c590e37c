"""A/B microbenchmark for the SpMM kernel variants on a Reddit-shaped graph.

Interleaved within-process rounds (guide §5.4 rule 24): variants alternate so
run-to-run variance is correlated. Reports ms and effective (logical) GB/s =
edges * F * 4 / t.

Usage (on a GPU box):
    python -m pipegcn_amd.tools.spmm_bench [--f 602 256] [--rounds 5]
"""
import argparse
import os
import time

import torch


def build(n=232_965, avg_deg=492, seed=0):
    from pipegcn_amd.graph.csr import HaloGraph
    g = torch.Generator().manual_seed(seed)
    d = torch.exp(torch.randn(n, generator=g)).clamp(min=0.1)
    deg = (d / d.mean() * avg_deg).round().long().clamp(min=1)
    v = torch.repeat_interleave(torch.arange(n), deg)
    u = (torch.rand(v.numel(), generator=g) * n).long()
    return HaloGraph.from_edges(u, v, n, n).to("cuda"), int(v.numel())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--f", type=int, nargs="+", default=[602, 256])
    ap.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32")
    ap.add_argument("--nodes", type=int, default=232_965)
    ap.add_argument("--deg", type=int, default=492)
    ap.add_argument("--rounds", type=int, default=5)
    ap.add_argument("--iters", type=int, default=3)
    args = ap.parse_args()

    from pipegcn_amd import ops

    hg, E = build(args.nodes, args.deg)
    n = hg.num_in
    deg = hg.csr.row_degrees().to("cuda").clamp(min=1)
    inv = (1.0 / deg).contiguous()
    print(f"graph: {n} nodes, {E} edges")

    elem = 2 if args.dtype == "bf16" else 4
    for F in args.f:
        feat = torch.randn(n, F, device="cuda")
        if args.dtype == "bf16":
            feat = feat.to(torch.bfloat16)
        variants = []
        for vec in (8, 4, 2, 1):
            if F % vec or vec * elem > 16:
                continue
            for order in ("o", "i"):
                variants.append((vec, order))
        results = {v: [] for v in variants}
        ref = None
        for rnd in range(args.rounds):
            for vec, order in variants:
                os.environ["PIPEGCN_SPMM_VEC"] = str(vec)
                os.environ["PIPEGCN_SPMM_ORDER"] = order
                out = ops.spmm(hg.csr, feat, inv)  # warm + correctness
                if ref is None:
                    ref = out.clone()
                else:
                    assert torch.allclose(out.float(), ref.float(),
                                          atol=1e-4), \
                        f"variant {vec}{order} WRONG"
                torch.cuda.synchronize()
                t0 = time.time()
                for _ in range(args.iters):
                    ops.spmm(hg.csr, feat, inv)
                torch.cuda.synchronize()
                results[(vec, order)].append((time.time() - t0) / args.iters)
        print(f"F={F}:")
        for (vec, order), ts in results.items():
            ms = min(ts) * 1e3
            gbs = E * F * elem / min(ts) / 1e9
            tag = "chunk-outer" if order == "o" else "chunk-inner"
            print(f"  VEC={vec} {tag:12s}: {ms:8.2f} ms  "
                  f"{gbs:8.0f} GB/s logical")
        os.environ.pop("PIPEGCN_SPMM_VEC")
        os.environ.pop("PIPEGCN_SPMM_ORDER")


if __name__ == "__main__":
    main()
