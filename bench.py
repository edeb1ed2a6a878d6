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
    ap.add_argument("--batch", type=int, default=16384,
                    help="per-rank batch (16384 measured +5.4%% over 8192 at 1 GPU "
                         "with the fused step, ops/s-neutral on the multi-rank "
                         "rehearsals after the round-2 protocol work)")
    ap.add_argument("--triples", type=int, default=90_000_000,
                    help="nominal epoch size (for epoch-time reporting)")
    ap.add_argument("--lookahead", type=int, default=4)
    ap.add_argument("--zipf", type=float, default=0.0,
                    help="entity skew exponent (0 = uniform, the default; e.g. "
                         "1.1 gives a Zipf-like hot set as in real KGs — hot "
                         "entities localize once instead of relocating per use)")
    ap.add_argument("--warmup-budget-s", type=float, default=4.0,
                    help="minimum untimed warmup wall-clock: after the W warmup "
                         "steps, keep running untimed steps until this budget is "
                         "spent AND step times stabilize (fresh boxes read "
                         "20-50%% slow for the first ~2s: DVFS/pager ramp)")
    ap.add_argument("--warmup-cap-steps", type=int, default=400,
                    help="hard cap on extra sustained-warmup steps")
    ap.add_argument("--no-intent", action="store_true",
                    help="disable intent signaling (pure remote-op mode)")
    ap.add_argument("--pipeline", type=int, default=0,
                    help="prefetch depth (bounded async, reference max_concurrent_loops); 0 = blocking per step")
    ap.add_argument("--no-fused", action="store_true",
                    help="disable the fused slab-direct kernel (single-rank GPU fast path)")
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--channels", type=int, default=4,
                    help="sync channels (key-space partitions; handlers run one "
                         "ordered unit per channel in parallel — 4 measured +5%% "
                         "over 2 on the 2-rank rehearsal)")
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

    def draw_entities(n):
        if args.zipf > 0:
            return (rng.zipf(args.zipf, size=n) - 1) % args.entities
        return rng.integers(0, args.entities, size=n)

    def make_batch(i):
        return np.stack([
            draw_entities(args.batch),
            rng.integers(0, args.relations, size=args.batch),
            draw_entities(args.batch),
        ], axis=1).astype(np.int64)

    total_steps = args.warmup + args.steps
    batches = [make_batch(i) for i in range(total_steps + args.lookahead)]

    def get_batch(i):
        while i >= len(batches):
            batches.append(make_batch(len(batches)))
        return batches[i]

    # optional prefetch pipeline (bounded async)
    from collections import deque

    handles = deque()

    # world==1: identity-layout fused kernel (zero host work); world>1:
    # general fused path (host offset pass + fused kernel on all-local
    # samples, classic path on the remote remainder)
    use_fused = (args.pipeline == 0) and (not args.no_fused) and is_cuda

    def issue(i):
        if not args.no_intent:
            model.signal_intent(get_batch(i + args.lookahead),
                                worker.current_clock() + args.lookahead,
                                worker.current_clock() + args.lookahead + args.pipeline + 2)
        handles.append(model.prefetch(get_batch(i)))

    if args.pipeline > 0:
        def run_step(i, sync_loss=False):
            issue(i + args.pipeline)
            loss = model.train_prefetched(handles.popleft(), sync_loss=sync_loss)
            worker.advance_clock()
            return loss

        for i in range(args.pipeline):
            issue(i)
    else:
        def run_step(i, sync_loss=False):
            if not args.no_intent:
                model.signal_intent(get_batch(i + args.lookahead),
                                    worker.current_clock() + args.lookahead,
                                    worker.current_clock() + args.lookahead + 2)
            if use_fused:
                loss = model.train_batch_fused(get_batch(i), sync_loss=sync_loss)
            else:
                loss = model.train_batch(get_batch(i), sync_loss=sync_loss)
            worker.advance_clock()
            return loss

    prof = None
    if os.environ.get("ADAPM_PROFILE") == "1":
        import cProfile
        prof = cProfile.Profile()

    keys_per_step = 2 * (3 * args.batch + args.batch * args.neg)  # pull + push

    # driver-contract warmup (the W untimed steps), timed for the cold rate
    t_cold = time.perf_counter()
    for i in range(args.warmup):
        run_step(i)
    model.drain()
    if is_cuda:
        torch.cuda.synchronize()
    cold_s = time.perf_counter() - t_cold
    cold_ops_per_s = keys_per_step * args.warmup * world / cold_s if args.warmup else None

    # sustained warmup (untimed): a fresh box ramps clocks/pager for ~2s
    # (measured 20-50% slow on the first steps) and, at world>1,
    # intent-driven relocation needs sync rounds to converge to >95%
    # locality — keep stepping until the budget is spent AND recent step
    # times are stable, so the K timed steps measure steady state.
    step_i = args.warmup
    extra = 0
    recent = []
    t_w = time.perf_counter()
    while extra < args.warmup_cap_steps:
        if time.perf_counter() - t_w >= args.warmup_budget_s and len(recent) >= 10:
            window = recent[-10:]
            if max(window) <= 1.25 * min(window):
                break
        t_s = time.perf_counter()
        run_step(step_i)
        if is_cuda:
            torch.cuda.synchronize()
        recent.append(time.perf_counter() - t_s)
        step_i += 1
        extra += 1
    model.drain()

    # classic-path reference sample (untimed region): the fused
    # slab-direct step is the world==1 hot path; the classic
    # pull/kernel/push path is what runs at world>1, so report both
    classic_ops_per_s = None
    if use_fused and world == 1:
        for _ in range(2):
            model.train_batch(get_batch(step_i))
            worker.advance_clock()
            step_i += 1
        torch.cuda.synchronize()
        t_c = time.perf_counter()
        for _ in range(6):
            model.train_batch(get_batch(step_i))
            worker.advance_clock()
            step_i += 1
        torch.cuda.synchronize()
        classic_ops_per_s = keys_per_step * 6 / (time.perf_counter() - t_c)

    if prof is not None:
        prof.enable()

    # timed region: EXACTLY args.steps steps, barrier+sync bracketed
    worker.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t_start = time.perf_counter()
    last_loss = 0.0
    for j in range(args.steps):
        last_loss = run_step(step_i + j, sync_loss=(j == args.steps - 1))
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

    ops_total = keys_per_step * args.steps * world
    ops_per_s = ops_total / t_elapsed
    triples_per_s = args.batch * args.steps * world / t_elapsed
    # estimate: nominal epoch size divided by the measured rate (an
    # actually-measured epoch needs --steps triples/batch/world)
    epoch_time_est_s = args.triples / triples_per_s
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
                "zipf": args.zipf,
                "fused_step": use_fused,
                "epoch_time_est_s": epoch_time_est_s,
                "triples_per_s": triples_per_s,
                "init_s": init_s,
                "warmup_extra_steps": extra,
                "cold_ops_per_s": cold_ops_per_s,
                "classic_ops_per_s": classic_ops_per_s,
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
