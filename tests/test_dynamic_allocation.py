"""Faithful mirror of reference tests/test_dynamic_allocation.cc:37-101:
all workers push {1,2} to ONE key fully async many times while random
intents relocate it around; the final value must be EXACTLY
world * runs * {1,2}."""
import numpy as np
import torch

from dist_helper import run_dist


def _hammer(rank, world):
    import adapm_amd

    adapm_amd.setup(num_keys=8, num_threads=1, device="cpu", max_sync_per_sec=4000.0)
    s = adapm_amd.Server(2)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    runs = 400
    key = np.array([3], dtype=np.int64)
    v = np.array([[1.0, 2.0]], dtype=np.float32)
    rng = np.random.default_rng(rank)
    for i in range(runs):
        if rng.random() < 0.1:  # random intent churn -> relocations mid-push
            w.intent(key, w.current_clock() + 1, w.current_clock() + int(rng.integers(2, 8)))
        w.push(key, v, async_=True)
        if i % 7 == 0:
            w.advance_clock()
    w.waitall()
    w.barrier()
    w.wait_sync()
    w.wait_sync()
    w.barrier()
    out = np.zeros((1, 2), dtype=np.float32)
    w.pull(key, out)
    exp = world * runs
    assert out[0, 0] == exp and out[0, 1] == 2 * exp, f"rank {rank}: {out} != {exp}"
    w.barrier()
    w.finalize()
    s.shutdown()


def test_dynamic_allocation_ws3():
    run_dist(3, _hammer, timeout=300)
