#!/usr/bin/env python3
"""CTR wide-and-deep benchmark (BASELINE config 5: 100M x 64 embedding
table; with --device-cap-gb the table exceeds the HBM budget and spills to
pinned host memory). Metric = examples/s."""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--features", type=int, default=100_000_000)
    ap.add_argument("--dim", type=int, default=64)
    ap.add_argument("--fields", type=int, default=16)
    ap.add_argument("--batch", type=int, default=16384)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--device-cap-gb", type=float, default=0.0,
                    help=">0: cap HBM arena; the rest spills to pinned host")
    ap.add_argument("--host-spill-gb", type=float, default=0.0)
    ap.add_argument("--hash-ids", action="store_true",
                    help="hash feature ids so hotness is uncorrelated with the "
                         "allocation order (realistic: ids are hashes) — otherwise "
                         "the Zipf-hot low ids happen to allocate into HBM first")
    ap.add_argument("--rebalance-every", type=int, default=0,
                    help=">0: call rebalance_spill every N steps (tiered store "
                         "keeps the hot set HBM-resident; world==1 only)")
    ap.add_argument("--rebalance-moves", type=int, default=131072)
    ap.add_argument("--rebalance-warmup-only", action="store_true",
                    help="stop rebalancing when the timed region starts (isolates "
                         "steady-state layout effect from rebalance call cost)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))

    import adapm_amd
    from adapm_amd.models.ctr import CTRConfig, WideAndDeep, make_synthetic_ctr

    cfg = CTRConfig(num_features=args.features, dim=args.dim, fields=args.fields,
                    batch_size=args.batch)
    adapm_amd.setup(num_keys=cfg.num_features, num_threads=1, device=args.device,
                    capacity_factor=1.3, max_sync_per_sec=2000.0,
                    device_cap_gb=args.device_cap_gb, host_spill_gb=args.host_spill_gb)
    server = adapm_amd.Server(cfg.row)
    worker = adapm_amd.Worker(0, server)
    model = WideAndDeep(cfg, server, worker)
    t_init0 = time.time()
    model.init_embeddings()
    init_s = time.time() - t_init0

    rng = np.random.default_rng(4000 + rank)
    total = args.warmup + args.steps
    batches = [make_synthetic_ctr(args.batch, args.features, args.fields, seed=1000 * rank + i)
               for i in range(total)]
    if args.hash_ids:
        batches = [((f * 2654435761) % args.features, y) for f, y in batches]
    do_rebalance = args.rebalance_every > 0 and world == 1
    rebal_stats = {"s": 0.0, "n": 0, "moves": 0}

    def maybe_rebalance(i):
        if args.rebalance_warmup_only and i >= args.warmup:
            return
        if do_rebalance and i % args.rebalance_every == 0:
            t = time.perf_counter()
            rebal_stats["moves"] += server.raw.rebalance_spill(args.rebalance_moves)
            rebal_stats["s"] += time.perf_counter() - t
            rebal_stats["n"] += 1

    is_cuda = server.rt.device.type == "cuda"
    for i in range(args.warmup):
        maybe_rebalance(i)
        model.train_batch(*batches[i], sync_loss=False)
    model.drain()
    worker.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.warmup, total):
        if i + 1 < total:
            model.signal_intent(batches[i + 1][0], worker.current_clock() + 1,
                                worker.current_clock() + 3)
        maybe_rebalance(i)
        model.train_batch(*batches[i], sync_loss=False)
        worker.advance_clock()
    model.drain()
    if is_cuda:
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    worker.barrier()
    if world > 1:
        el = worker.allreduce(el, op="max")

    st = server.stats()
    if rank == 0 and rebal_stats["n"]:
        print(f"rebalance: {rebal_stats['n']} calls, {rebal_stats['s']:.3f}s total, "
              f"{1000*rebal_stats['s']/rebal_stats['n']:.1f} ms/call, {rebal_stats['moves']} moves",
              file=sys.stderr, flush=True)
    if rank == 0 and os.environ.get("ADAPM_CPP_TIMING"):
        print("cpp timing:", {k: v for k, v in st.items() if k.startswith("t_")},
              file=sys.stderr, flush=True)
    if rank == 0:
        print(json.dumps({
            "metric": "ctr_examples_per_s", "value": args.batch * args.steps * world / el,
            "unit": "examples/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup, "ms_per_step": 1000 * el / args.steps,
            "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
            "dtype": "fp32", "data": "synthetic",
            "config": {"model": "ctr_wide_and_deep_100Mx64", "features": args.features,
                       "dim": args.dim, "fields": args.fields,
                       "global_batch": args.batch * world, "init_s": init_s,
                       "device_cap_gb": args.device_cap_gb,
                       "host_spill_in_use_gb": st["host_spill_in_use"] * 4 / (1 << 30),
                       "spill_rebalance_moves": st.get("spill_rebalance_moves", 0),
                       "hash_ids": args.hash_ids,
                       "pull_push_ops_per_s": 2 * args.batch * args.fields * args.steps * world / el,
                       "parallelism": f"ps-async-dp{world}"},
        }), flush=True)
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
