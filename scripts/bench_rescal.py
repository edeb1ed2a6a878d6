#!/usr/bin/env python3
"""RESCAL step kernel A/B: classic per-triple scalar kernel
(_C.rescal_step, LDS-staged R, VALU matvecs) vs the MFMA grouped path
(_C.rescal_step_grouped: U/dS/dR as matrix-core GEMMs over
relation-sorted groups). VERDICT r01 item 8 asks for >=2x at D=128.

Run on a GPU box:  python scripts/bench_rescal.py [--dim 128] [--batch 2048]
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--batch", type=int, default=2048)
    ap.add_argument("--relations", type=int, default=32)
    ap.add_argument("--neg", type=int, default=8)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    args = ap.parse_args()
    from adapm_amd import _C

    B, D, G, N = args.batch, args.dim, args.relations, args.neg
    assert torch.cuda.is_available()
    gen = torch.Generator().manual_seed(0)
    rels = np.sort(np.random.default_rng(0).integers(0, G, size=B))
    uniq, first = np.unique(rels, return_index=True)
    starts = np.append(first, B).astype(np.int32)
    Gu = len(uniq)

    s = torch.randn(B, 2 * D, generator=gen).abs_().cuda()
    o = torch.randn(B, 2 * D, generator=gen).abs_().cuda()
    neg = torch.randn(B * N, 2 * D, generator=gen).abs_().cuda()
    rm_u = torch.randn(Gu, 2 * D * D, generator=gen).abs_().mul_(0.05).cuda()
    # classic path pulls one R copy per triple
    rep = torch.from_numpy(np.searchsorted(uniq, rels)).cuda()
    rm_b = rm_u.index_select(0, rep).contiguous()

    ds = torch.empty_like(s)
    do = torch.empty_like(o)
    dn = torch.empty_like(neg)
    drl_b = torch.empty_like(rm_b)
    drl_u = torch.empty_like(rm_u)
    loss = torch.empty(B, device="cuda")
    starts_t = torch.from_numpy(starts)

    def run_classic():
        _C.rescal_step(s, rm_b, o, neg, ds, drl_b, do, dn, loss, N, D, 0.1, 1e-8)

    def run_grouped():
        _C.rescal_step_grouped(s, rm_u, o, neg, ds, drl_u, do, dn, starts_t, N, D, 0.1, 1e-8)

    out = {}
    for name, fn in (("classic", run_classic), ("grouped_mfma", run_grouped)):
        for _ in range(args.warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        out[name] = dt
        print(f"{name}: {dt*1e3:.3f} ms/step  "
              f"({B / dt:.0f} triples/s, D={D}, G={Gu}, N={N})")
    print(f"speedup: {out['classic'] / out['grouped_mfma']:.2f}x "
          f"(classic also pulls {B}x R rows vs {Gu}x for grouped — "
          f"pull/push savings not counted here)")


if __name__ == "__main__":
    main()
