"""Set (overwrite) semantics interleaved with push/pull (mirrors reference
tests/test_set_operation.cc)."""
import numpy as np
import torch

from dist_helper import run_dist


def _set_ops(rank, world):
    import adapm_amd

    adapm_amd.setup(num_keys=16, num_threads=1, device="cpu", max_sync_per_sec=2000.0)
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    w.barrier()

    # each rank owns distinct keys; set on own key then everyone reads
    my_key = np.array([rank], dtype=np.int64)
    w.set(my_key, np.full((1, 4), 10.0 * (rank + 1), dtype=np.float32))
    w.wait_sync()
    w.barrier()
    for r in range(world):
        out = np.zeros((1, 4), dtype=np.float32)
        w.pull(np.array([r], dtype=np.int64), out)
        assert np.allclose(out, 10.0 * (r + 1)), f"rank {rank} read {out} for key {r}"
    w.barrier()

    # set overwrites prior pushes (on the owner)
    w.push(my_key, np.ones((1, 4), dtype=np.float32))
    w.set(my_key, np.full((1, 4), 5.0, dtype=np.float32))
    out = np.zeros((1, 4), dtype=np.float32)
    w.pull(my_key, out)
    assert np.allclose(out, 5.0)
    # push after set is additive on top
    w.push(my_key, np.ones((1, 4), dtype=np.float32))
    w.pull(my_key, out)
    assert np.allclose(out, 6.0)
    w.barrier()

    # remote set: rank 0 overwrites a key owned by rank (world-1)
    tgt = np.array([world - 1], dtype=np.int64)
    if rank == 0:
        w.set(tgt, np.full((1, 4), 77.0, dtype=np.float32))
    w.wait_sync()
    w.barrier()
    out = np.zeros((1, 4), dtype=np.float32)
    w.pull(tgt, out)
    assert np.allclose(out, 77.0), f"rank {rank}: {out}"
    w.barrier()
    w.finalize()
    s.shutdown()


def test_set_operation_ws4():
    run_dist(4, _set_ops, timeout=240)
