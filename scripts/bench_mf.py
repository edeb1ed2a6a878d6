#!/usr/bin/env python3
"""Matrix-factorization benchmark (BASELINE config 3: rank=128, 10M x 1M
synthetic ratings). Metric = nonzeros (rating updates) per second."""
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
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--cols", type=int, default=1_000_000)
    ap.add_argument("--rank", type=int, default=128)
    ap.add_argument("--batch-nnz", type=int, default=131072)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--no-fused", action="store_true",
                    help="disable the fused slab-direct kernel (single-rank GPU fast path)")
    args = ap.parse_args()

    rank_id = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))

    import adapm_amd
    from adapm_amd.models.mf import MF, MFConfig

    cfg = MFConfig(num_rows=args.rows, num_cols=args.cols, rank=args.rank,
                   batch_nnz=args.batch_nnz)
    adapm_amd.setup(num_keys=cfg.num_keys, num_threads=1, device=args.device,
                    capacity_factor=2.0, max_sync_per_sec=2000.0)
    server = adapm_amd.Server(cfg.row)
    worker = adapm_amd.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()

    rng = np.random.default_rng(3000 + rank_id)
    total = args.warmup + args.steps
    batches = [(rng.integers(0, args.rows, args.batch_nnz),
                rng.integers(0, args.cols, args.batch_nnz),
                rng.normal(size=args.batch_nnz).astype(np.float32)) for _ in range(total)]

    is_cuda = server.rt.device.type == "cuda"
    use_fused = (not args.no_fused) and world == 1 and is_cuda
    step_fn = model.train_batch_fused if use_fused else model.train_batch
    for i in range(args.warmup):
        step_fn(*batches[i])
    model.drain()
    worker.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.warmup, total):
        # intent one batch ahead (plain_sgd schedule shape)
        if i + 1 < total:
            r2, c2, _ = batches[i + 1]
            worker.intent(np.concatenate([r2.astype(np.int64), model.col_key(c2)]),
                          worker.current_clock() + 1, worker.current_clock() + 3)
        step_fn(*batches[i])
        worker.advance_clock()
    model.drain()
    if is_cuda:
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    worker.barrier()
    if world > 1:
        el = worker.allreduce(el, op="max")

    nnz_per_s = args.batch_nnz * args.steps * world / el
    if rank_id == 0:
        print(json.dumps({
            "metric": "mf_nnz_updates_per_s", "value": nnz_per_s, "unit": "nnz/s",
            "n_gpus": world, "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": 1000 * el / args.steps, "higher_is_better": True,
            "scaling": "weak", "vs_baseline": None, "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": "mf_rank128", "rows": args.rows, "cols": args.cols,
                       "rank": args.rank, "global_batch": args.batch_nnz * world,
                       "pull_push_ops_per_s": 4 * args.batch_nnz * args.steps * world / el,
                       "parallelism": f"ps-async-dp{world}"},
        }), flush=True)
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
