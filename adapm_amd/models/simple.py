"""Smoke-test app: per-iteration Intent -> Push -> Pull over a tiny key
space (rebuild of reference apps/simple.cc:36-67)."""
from __future__ import annotations

import numpy as np
import torch

import adapm_amd


def run_simple(iterations: int = 50, num_keys: int = 1000, vpk: int = 10,
               device: str = "cpu", verbose: bool = False):
    adapm_amd.setup(num_keys=num_keys, num_threads=1, device=device)
    server = adapm_amd.Server(vpk)
    worker = adapm_amd.Worker(0, server)
    rank = server.my_rank()
    worker.barrier()
    rng = np.random.default_rng(rank)
    for it in range(iterations):
        keys = rng.choice(num_keys, size=3, replace=False).astype(np.int64)
        worker.intent(keys, worker.current_clock() + 1, worker.current_clock() + 3)
        vals = np.ones((3, vpk), dtype=np.float32)
        worker.push(keys, vals)
        out = np.zeros((3, vpk), dtype=np.float32)
        worker.pull(keys, out)
        assert (out >= vals - 1e-3).all(), f"iteration {it}: pulled {out}"
        worker.advance_clock()
        if verbose and it % 10 == 0:
            print(f"[simple] rank {rank} iteration {it} ok", flush=True)
    worker.barrier()
    total = worker.allreduce(float(iterations))
    worker.finalize()
    stats = server.stats()
    server.shutdown()
    if verbose:
        print(f"[simple] rank {rank} done; global iterations {total}; stats {stats}")
    return stats


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--iterations", type=int, default=50)
    ap.add_argument("--num-keys", type=int, default=1000)
    ap.add_argument("--device", type=str, default="cpu")
    a = ap.parse_args()
    run_simple(a.iterations, a.num_keys, device=a.device, verbose=True)
