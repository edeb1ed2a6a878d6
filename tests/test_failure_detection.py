"""Failure surfacing: when a peer dies mid-run, blocked Wait()s return and
the failure is reported instead of hanging (reference heartbeat/dead-node
scope, van.cc:515-527 + tracker keepalive)."""
import os
import sys
import time

import numpy as np

from dist_helper import run_dist


def _victim_or_survivor(rank, world):
    import adapm_amd

    adapm_amd.setup(num_keys=32, num_threads=1, device="cpu", max_sync_per_sec=2000.0)
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    if rank == 1:
        os._exit(1)  # simulate a node crash
    # survivor: issue a remote op toward the dead rank; must NOT hang
    t0 = time.monotonic()
    out = np.zeros((1, 4), dtype=np.float32)
    try:
        w.pull(np.array([1], dtype=np.int64), out)  # key 1 owned by dead rank
    except Exception:
        pass
    deadline = time.monotonic() + 120
    while not s._sync.failed and time.monotonic() < deadline:
        time.sleep(0.2)
    assert s._sync.failed, "sync failure never detected"
    assert s.raw.failed_reason() != ""
    elapsed = time.monotonic() - t0
    assert elapsed < 110, f"took {elapsed}s to detect failure"
    os._exit(0)  # skip normal shutdown (collective would hang)


def test_peer_death_detected():
    # custom runner: the victim's deliberate death confuses run_dist's
    # accounting, so check the survivor's exit code directly
    import multiprocessing as mp

    from dist_helper import _entry, _free_port

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_entry, args=(r, 2, port, _victim_or_survivor, (), q),
                         daemon=True)
             for r in range(2)]
    for p in procs:
        p.start()
    procs[0].join(timeout=150)  # the survivor (rank 0)
    alive = procs[0].is_alive()
    code = procs[0].exitcode
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert not alive, "survivor hung after peer death"
    assert code == 0, f"survivor exited {code}"

