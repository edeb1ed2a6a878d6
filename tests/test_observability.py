"""Locality stats + per-key trace TSV outputs (reference PS_LOCALITY_STATS
/ PS_TRACE_KEYS, coloc_kv_server_handle.h:86-118, 960-992)."""
import os

import numpy as np
import torch

from dist_helper import run_dist


def test_locality_stats_tsv(tmp_path):
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    out = str(tmp_path)
    adapm_amd.setup(num_keys=32, num_threads=1, device="cpu", locality_stats=True,
                    trace_keys="all", stats_out=out)
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    keys = np.array([1, 2, 3], dtype=np.int64)
    w.push(keys, np.ones((3, 4), dtype=np.float32))
    o = np.zeros((3, 4), dtype=np.float32)
    w.pull(keys, o)
    w.pull(keys, o)
    s.shutdown()

    ls = os.path.join(out, "locality_stats.rank.0.tsv")
    assert os.path.exists(ls)
    lines = open(ls).read().splitlines()
    assert lines[0] == "key\taccesses\tlocal"
    rows = {int(l.split("\t")[0]): l.split("\t") for l in lines[1:]}
    # 1-rank fast path counts only pulls per key in the slow path; the
    # counters are on the tracked (slow) path — with world==1 the fast
    # path is used, so per-key counters may be empty. Force slow path:
    # (covered by the distributed test below instead)
    assert os.path.exists(os.path.join(out, "traces.0.tsv"))


def _dist_obs(rank, world, out):
    import adapm_amd

    adapm_amd.setup(num_keys=32, num_threads=1, device="cpu", locality_stats=True,
                    trace_keys="all", stats_out=out, max_sync_per_sec=2000.0)
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    # force intent-driven transitions so traces have events
    w.intent(np.array([0, 1], dtype=np.int64), 1, 100)
    import time

    time.sleep(0.3)
    o = np.zeros((2, 4), dtype=np.float32)
    w.pull(np.array([0, 1], dtype=np.int64), o)
    # a remote pull so hop/remote-served stats have data (key 2+world is
    # owned by the other rank and has no intent signaled)
    w.pull(np.array([2 + (1 - rank), 4 + rank], dtype=np.int64),
           np.zeros((2, 4), dtype=np.float32))
    w.barrier()
    w.wait_sync(strong=True)
    st = s.stats()
    # hop histogram (reference sync_manager.h hop stats): every served
    # remote op lands in a bucket; mean hops ~0 in a 2-rank run
    assert sum(st["hop_hist"]) == st["remote_pulls_served"] + st["remote_pushes_served"]
    assert st["replica_records"] >= st["replica_payloads"] >= 0
    w.barrier()
    w.finalize()
    s.shutdown()
    tr = os.path.join(out, f"traces.{rank}.tsv")
    assert os.path.exists(tr)
    txt = open(tr).read()
    if rank != 0:
        assert "REPLICA_SETUP" in txt or "RELOC_IN" in txt, txt[:300]
    ls = open(os.path.join(out, f"locality_stats.rank.{rank}.tsv")).read()
    assert ls.startswith("key\taccesses\tlocal")


def test_observability_dist(tmp_path):
    run_dist(2, _dist_obs, str(tmp_path), timeout=180)
