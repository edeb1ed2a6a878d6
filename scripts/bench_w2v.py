#!/usr/bin/env python3
"""word2vec SGNS benchmark (BASELINE config 2: dim=300, 1M-vocab synthetic
corpus, 1x MI355X). Same JSON contract as bench.py; metric = SGNS
(center, context) pairs trained per second, whole job."""
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
    ap.add_argument("--vocab", type=int, default=1_000_000)
    ap.add_argument("--dim", type=int, default=300)
    ap.add_argument("--negative", type=int, default=5)
    ap.add_argument("--pairs", type=int, default=65536, help="pairs per step")
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--no-fused", action="store_true",
                    help="disable the fused slab-direct kernel (single-rank GPU fast path)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))

    import adapm_amd
    from adapm_amd.models.word2vec import W2VConfig, Word2Vec

    cfg = W2VConfig(vocab_size=args.vocab, dim=args.dim, negative=args.negative)
    adapm_amd.setup(num_keys=cfg.num_keys, num_threads=1, device=args.device,
                    capacity_factor=2.0, max_sync_per_sec=2000.0)
    server = adapm_amd.Server(cfg.row)
    counts = (1.0 / np.arange(1, args.vocab + 1)) ** 0.7 * 1e9  # Zipf-ish counts
    server.enable_sampling_support("local", True, "unigram", 0, args.vocab, counts=counts)
    worker = adapm_amd.Worker(0, server)
    model = Word2Vec(cfg, server, worker)
    model.set_vocab_counts(counts)
    model.init_embeddings()

    rng = np.random.default_rng(2000 + rank)
    # training pairs follow the SUBSAMPLED word distribution (reference
    # word2vec 'sample' parameter drops frequent words before pairing —
    # the same subsampling Word2Vec.set_vocab_counts implements), drawn
    # via the alias table
    from adapm_amd.sampling import Unigram

    f = counts / counts.sum()
    keep = np.minimum(1.0, np.sqrt(1e-3 / f) + 1e-3 / f)
    pair_dist = Unigram(counts * keep, None, 1.0, seed=2000 + rank,
                        device=server.rt.device)
    total = args.warmup + args.steps
    batches = [(pair_dist.draw(args.pairs), pair_dist.draw(args.pairs))
               for _ in range(total)]

    is_cuda = server.rt.device.type == "cuda"
    use_fused = (not args.no_fused) and world == 1 and is_cuda
    step_fn = model.train_pairs_fused if use_fused else model.train_pairs
    for i in range(args.warmup):
        step_fn(*batches[i])
    model.drain()
    worker.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.warmup, total):
        step_fn(*batches[i])
    model.drain()
    if is_cuda:
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    worker.barrier()
    if world > 1:
        el = worker.allreduce(el, op="max")

    pairs_per_s = args.pairs * args.steps * world / el
    keys_per_step = 2 * (2 + args.negative) * args.pairs
    if rank == 0:
        print(json.dumps({
            "metric": "sgns_pairs_per_s", "value": pairs_per_s, "unit": "pairs/s",
            "n_gpus": world, "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": 1000 * el / args.steps, "higher_is_better": True,
            "scaling": "weak", "vs_baseline": None, "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": "word2vec_sgns_dim300", "vocab": args.vocab,
                       "dim": args.dim, "negative": args.negative,
                       "global_batch": args.pairs * world,
                       "pull_push_ops_per_s": keys_per_step * args.steps * world / el,
                       "parallelism": f"ps-async-dp{world}"},
        }), flush=True)
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
