"""Multiple worker threads per process (the reference's N worker threads
per node, README.md:161-165): concurrent ops from Python threads, shared
server, per-worker clocks, thread-group barrier."""
import threading

import numpy as np
import torch

from dist_helper import run_dist


def test_two_workers_single_rank():
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=64, num_threads=2, device="cpu")
    s = adapm_amd.Server(4)
    results = []

    def run(wid):
        w = adapm_amd.Worker(wid, s)
        for it in range(200):
            keys = np.array([wid, 32 + it % 8], dtype=np.int64)
            w.push(keys, np.ones((2, 4), dtype=np.float32), async_=True)
            w.advance_clock()
        w.waitall()
        w.barrier()
        out = np.zeros((1, 4), dtype=np.float32)
        w.pull(np.array([32], dtype=np.int64), out)
        results.append(out[0, 0])
        w.barrier()

    ts = [threading.Thread(target=run, args=(i,)) for i in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=60)
        assert not t.is_alive()
    # both workers pushed 25 times to key 32 (it%8==0) => 50 total
    assert all(abs(r - 50.0) < 1e-3 for r in results), results
    s.shutdown()


def _multi_worker_dist(rank, world):
    import adapm_amd

    adapm_amd.setup(num_keys=48, num_threads=2, device="cpu", max_sync_per_sec=2000.0)
    s = adapm_amd.Server(2)
    errs = []

    def run(wid):
        try:
            w = adapm_amd.Worker(wid, s)
            w.barrier()
            rng = np.random.default_rng(rank * 10 + wid)
            n_push = np.zeros(48)
            for it in range(100):
                keys = rng.choice(48, size=2, replace=False).astype(np.int64)
                if rng.random() < 0.4:
                    w.intent(keys, w.current_clock() + 1, w.current_clock() + 10)
                w.push(keys, np.ones((2, 2), dtype=np.float32), async_=True)
                n_push[keys] += 1
                w.advance_clock()
            w.waitall()
            w.barrier()
            w.wait_sync()
            w.wait_sync()
            w.barrier()
            total = w.allreduce(torch.tensor(n_push, dtype=torch.float32)) if wid == 0 else None
            if wid == 0:
                out = np.zeros((48, 2), dtype=np.float32)
                w.pull(np.arange(48, dtype=np.int64), out)
                # local allreduce covers only worker 0's pushes of each rank;
                # add worker 1's via a second exchange is overkill — just
                # check monotonic lower bound
                assert (out[:, 0] >= total.numpy() - 1e-2).all()
            w.barrier()
        except Exception as e:  # pragma: no cover
            errs.append(e)
            raise

    ts = [threading.Thread(target=run, args=(i,)) for i in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=120)
    assert not errs and all(not t.is_alive() for t in ts)
    s.shutdown()


def test_multi_worker_ws2():
    run_dist(2, _multi_worker_dist, timeout=300)


def test_concurrent_slow_pass_threads():
    """Concurrent worker threads on a NON-identity store (spill forces the
    slow metadata pass) — the PassPool must serialize its task slot or a
    caller hangs."""
    import threading

    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    N = 50_000
    adapm_amd.setup(num_keys=N, num_threads=4, device="cpu",
                    device_cap_gb=N * 32 * 4 * 0.5 / 2**30, host_spill_gb=0.5)
    s = adapm_amd.Server(32)
    workers = [adapm_amd.Worker(i, s) for i in range(4)]
    vals = torch.ones(2000, 32)
    errs = []

    def hammer(w, seed):
        try:
            rng = np.random.default_rng(seed)
            for _ in range(25):
                ks = rng.choice(N, 2000, replace=False).astype(np.int64)
                w.push(ks, vals)
                out = torch.zeros(2000, 32)
                w.pull(ks, out)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=hammer, args=(w, i)) for i, w in enumerate(workers)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(90)
    assert not errs, errs
    assert not any(t.is_alive() for t in ts), "a worker hung in the slow pass"
    s.shutdown()
    adapm_amd._SETUP.clear()
