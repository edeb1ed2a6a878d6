#!/usr/bin/env python3
"""Flagship benchmark: KGE ComplEx dim-512 training on the adaptive
parameter manager (BASELINE.json north-star config: "Pull+Push ops/sec
(whole node) + epoch time, KGE ComplEx dim-512, 90M-triple synthetic
graph").

Run:  python bench.py --gpus N --steps K --warmup W
Multi-GPU: launched by the driver via torch.distributed.run (one rank per
GPU over RCCL); reads RANK/WORLD_SIZE/MASTER_* from the env.

A "step" = one training batch: pull (s, r, o, negatives) -> fused ComplEx
score/grad/AdaGrad kernel -> push deltas, with intent signaled
`lookahead` batches ahead. The metric is Pull+Push key-ops/sec summed
over all ranks (each pulled key + each pushed key = one op), on synthetic
triples and random-init embeddings (no network for datasets).
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--entities", type=int, default=10_000_000)
    ap.add_argument("--relations", type=int, default=1_000)
    ap.add_argument("--dim", type=int, default=512)
    ap.add_argument("--neg", type=int, default=16)
    ap.add_argument("--batch", type=int, default=8192)
    ap.add_argument("--triples", type=int, default=90_000_000,
                    help="nominal epoch size (for epoch-time reporting)")
    ap.add_argument("--lookahead", type=int, default=4)
    ap.add_argument("--no-intent", action="store_true",
                    help="disable intent signaling (pure remote-op mode)")
    ap.add_argument("--pipeline", type=int, default=0,
                    help="prefetch depth (bounded async, reference max_concurrent_loops); 0 = blocking per step")
    ap.add_argument("--no-fused", action="store_true",
                    help="disable the fused slab-direct kernel (single-rank GPU fast path)")
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--channels", type=int, default=2)
    ap.add_argument("--capacity-factor", type=float, default=3.0)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    os.environ.setdefault("RANK", str(rank))
    os.environ.setdefault("WORLD_SIZE", str(world))

    import adapm_amd
    from adapm_amd.models.kge import ComplEx, ComplExConfig, make_synthetic_triples

    adapm_amd.setup(num_keys=args.entities + args.relations, num_threads=1,
                    num_channels=args.channels, device=args.device,
                    capacity_factor=args.capacity_factor,
                    max_sync_per_sec=2000.0)
    server = adapm_amd.Server(2 * args.dim)
    server.enable_sampling_support("local", True, "uniform", 0, args.entities)
    worker = adapm_amd.Worker(0, server)

    cfg = ComplExConfig(num_entities=args.entities, num_relations=args.relations,
                        dim=args.dim, neg_samples=args.neg, batch_size=args.batch,
                        lookahead=args.lookahead, seed=7)
    model = ComplEx(cfg, server, worker)
    t0 = time.time()
    model.init_embeddings()
    init_s = time.time() - t0

    dev = server.rt.device
    is_cuda = dev.type == "cuda"
    rng = np.random.default_rng(1000 + rank)

    def make_batch(i):
        return np.stack([
            rng.integers(0, args.entities, size=args.batch),
            rng.integers(0, args.relations, size=args.batch),
            rng.integers(0, args.entities, size=args.batch),
        ], axis=1).astype(np.int64)

    total_steps = args.warmup + args.steps
    batches = [make_batch(i) for i in range(total_steps + args.lookahead)]

    # optional prefetch pipeline (bounded async)
    from collections import deque

    handles = deque()

    def issue(i):
        if not args.no_intent:
            model.signal_intent(batches[i + args.lookahead],
                                worker.current_clock() + args.lookahead,
                                worker.current_clock() + args.lookahead + args.pipeline + 2)
        handles.append(model.prefetch(batches[i]))

    if args.pipeline > 0:
        def run_step(i, sync_loss=False):
            if i + args.pipeline < total_steps:
                issue(i + args.pipeline)
            loss = model.train_prefetched(handles.popleft(), sync_loss=sync_loss)
            worker.advance_clock()
            return loss

        for i in range(min(args.pipeline, total_steps)):
            issue(i)
    else:
        use_fused = (not args.no_fused) and world == 1 and is_cuda

        def run_step(i, sync_loss=False):
            if not args.no_intent:
                model.signal_intent(batches[i + args.lookahead],
                                    worker.current_clock() + args.lookahead,
                                    worker.current_clock() + args.lookahead + 2)
            if use_fused:
                loss = model.train_batch_fused(batches[i], sync_loss=sync_loss)
            else:
                loss = model.train_batch(batches[i], sync_loss=sync_loss)
            worker.advance_clock()
            return loss

    prof = None
    if os.environ.get("ADAPM_PROFILE") == "1":
        import cProfile
        prof = cProfile.Profile()

    # warmup
    for i in range(args.warmup):
        run_step(i)
    model.drain()
    if prof is not None:
        prof.enable()

    # timed region
    worker.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t_start = time.perf_counter()
    last_loss = 0.0
    for i in range(args.warmup, total_steps):
        last_loss = run_step(i, sync_loss=(i == total_steps - 1))
    model.drain()
    if is_cuda:
        torch.cuda.synchronize()
    t_elapsed = time.perf_counter() - t_start
    if prof is not None:
        prof.disable()
        import pstats
        pstats.Stats(prof, stream=sys.stderr).sort_stats("cumulative").print_stats(25)
    worker.barrier()

    # MAX elapsed over ranks
    if world > 1:
        t_elapsed = worker.allreduce(t_elapsed, op="max")

    keys_per_step = 2 * (3 * args.batch + args.batch * args.neg)  # pull + push
    ops_total = keys_per_step * args.steps * world
    ops_per_s = ops_total / t_elapsed
    triples_per_s = args.batch * args.steps * world / t_elapsed
    epoch_time_s = args.triples / triples_per_s
    ms_per_step = 1000.0 * t_elapsed / args.steps

    st = server.stats()
    if rank == 0 and os.environ.get("ADAPM_CPP_TIMING"):
        print("cpp timing:", {k: v for k, v in st.items() if k.startswith("t_")},
              file=sys.stderr, flush=True)
    if rank == 0 and model.phase_times:
        print("phase times (s over timed+warmup):",
              dict(sorted(model.phase_times.items())), file=sys.stderr, flush=True)
    if rank == 0:
        out = {
            "metric": "pull_push_ops_per_s",
            "value": ops_per_s,
            "unit": "key-ops/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "kge_complex_dim512",
                "entities": args.entities,
                "relations": args.relations,
                "dim": args.dim,
                "neg_samples": args.neg,
                "global_batch": args.batch * world,
                "triples": args.triples,
                "parallelism": f"ps-async-dp{world}",
                "fused_step": (not args.no_fused) and world == 1 and dev.type == "cuda",
                "epoch_time_s": epoch_time_s,
                "triples_per_s": triples_per_s,
                "init_s": init_s,
                "last_loss": last_loss,
                "pull_local_frac": st["pull_local"] / max(1, st["pull_keys"]),
                "push_local_frac": st["push_local"] / max(1, st["push_keys"]),
            },
        }
        print(json.dumps(out), flush=True)

    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
