"""Failure propagation: hop-limit give-ups must fail loudly (NACK -> the
origin's wait() raises) instead of acking a dropped op, and ops issued
after a transport failure raise instead of creating tickets that can
never complete. (The reference never drops — its addressbook routing
always converges at the manager, addressbook.h:92-112 — so any drop here
is an error condition that must surface.)"""
import os

import numpy as np
import pytest
import torch

import adapm_amd
from dist_helper import run_dist


def test_ops_raise_after_fail():
    adapm_amd.setup(num_keys=16, num_threads=1, device="cpu")
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    w.push([1], torch.ones(1, 4))
    s.raw.fail("injected test failure")
    with pytest.raises(RuntimeError, match="injected test failure"):
        w.pull([1], torch.zeros(1, 4))
    with pytest.raises(RuntimeError, match="injected test failure"):
        w.push([1], torch.ones(1, 4))
    # wait on a stale ts returns/raises, never hangs
    with pytest.raises(RuntimeError):
        s.raw.wait_rounds([10**9])
    s._shut = True  # skip collective shutdown (world=1: nothing to do)
    from adapm_amd import runtime as _rt

    _rt.shutdown_runtime()


def _hop_limit_worker(rank, world, kind):
    os.environ["ADAPM_MAX_HOPS"] = "0"
    import adapm_amd as A

    A.setup(num_keys=8, num_threads=1, device="cpu", max_sync_per_sec=5000.0)
    s = A.Server(4)
    w = A.Worker(0, s)
    w.barrier()
    raised = False
    if rank == 1:
        # poison the location cache: key 0 (owned by rank 0) appears to
        # live on rank 1 itself -> the request is handled locally, we are
        # not the owner, and with max_hops=0 the forward NACKs immediately.
        s.raw.debug_set_loc_cache(0, 1)
        try:
            if kind == "pull":
                w.pull(np.array([0], dtype=np.int64), np.zeros((1, 4), dtype=np.float32))
            else:
                w.push(np.array([0], dtype=np.int64), np.ones((1, 4), dtype=np.float32))
        except RuntimeError as e:
            assert "hop limit" in str(e), str(e)
            raised = True
        assert raised, f"{kind} at the hop cap completed as success (silent drop)"
    w.barrier()
    s.shutdown()


@pytest.mark.parametrize("kind", ["pull", "push"])
def test_hop_limit_fails_loudly(kind):
    run_dist(2, _hop_limit_worker, kind)
