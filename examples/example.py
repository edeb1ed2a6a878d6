"""API walkthrough (equivalent of reference bindings/example.py): multiple
ranks x multiple worker threads, torch + numpy tensors, intent, sampling.

Run:  python -m adapm_amd.launch -n 4 examples/example.py
"""
import os
import sys
import threading

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import adapm_amd

NUM_KEYS = 1000
LEN = 16
WORKERS = 2


def run_worker(wid, server):
    worker = adapm_amd.Worker(wid, server)
    worker.barrier()
    rank = server.my_rank()

    # torch tensors
    keys = torch.tensor([1, 2, 3])
    vals = torch.ones(3, LEN)
    worker.push(keys, vals)
    out = torch.zeros(3, LEN)
    worker.pull(keys, out)

    # numpy + async
    nkeys = np.array([10, 11], dtype=np.int64)
    ts = worker.push(nkeys, np.full((2, LEN), 2.0, dtype=np.float32), async_=True)
    worker.wait(ts)

    # intent-driven localization
    hot = np.array([42 + rank], dtype=np.int64)
    worker.intent(hot, worker.current_clock() + 1, worker.current_clock() + 1000)
    worker.wait_sync()
    o = np.zeros((1, LEN), dtype=np.float32)
    t = worker.pull(hot, o, async_=True)
    print(f"[rank {rank} worker {wid}] hot-key pull ts={t} (-1 means local)")
    worker.wait(t)

    # sampling
    sid = worker.prepare_sample(8, worker.current_clock(), worker.current_clock() + 10)
    skeys = np.zeros(8, dtype=np.int64)
    svals = np.zeros((8, LEN), dtype=np.float32)
    worker.pull_sample(sid, skeys, svals)
    worker.finish_sample(sid)

    # loss aggregation
    total = worker.allreduce(float(rank))
    if wid == 0 and rank == 0:
        print(f"allreduce(ranks) = {total}")

    worker.barrier()
    worker.finalize()


def main():
    adapm_amd.setup(num_keys=NUM_KEYS, num_threads=WORKERS)
    server = adapm_amd.Server(LEN)
    server.enable_sampling_support("local", True, "uniform", 0, NUM_KEYS)
    threads = [threading.Thread(target=run_worker, args=(w, server)) for w in range(WORKERS)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    print(f"[rank {server.my_rank()}] stats: pulls={server.stats()['pull_keys']}")
    server.shutdown()


if __name__ == "__main__":
    main()
