"""Coverage for less-traveled paths: non-uniform lengths across ranks
(exercises the per-key record path — bulk records need uniform stores),
4-channel configuration, DMLC_* env fallbacks."""
import os

import numpy as np
import pytest
import torch

from dist_helper import run_dist


def _non_uniform_dist(rank, world):
    import adapm_amd

    NUM = 40
    adapm_amd.setup(num_keys=NUM, num_threads=1, device="cpu", max_sync_per_sec=2000.0)
    lens = torch.tensor([(2 + (k % 3) * 4) for k in range(NUM)])  # 2/6/10 floats
    s = adapm_amd.Server(lens)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    # remote pushes + pulls over per-key records
    for k in range(NUM):
        l = int(lens[k])
        w.push(np.array([k], dtype=np.int64), np.full((1, l), 1.0, dtype=np.float32))
    w.barrier()
    w.wait_sync()
    for k in range(NUM):
        l = int(lens[k])
        out = np.zeros((1, l), dtype=np.float32)
        w.pull(np.array([k], dtype=np.int64), out)
        assert np.allclose(out, float(world)), f"key {k}: {out}"
    # intents on a non-uniform store (replication path with varying lens)
    w.intent(np.arange(0, NUM, 5, dtype=np.int64), 1, 1_000_000)
    import time

    time.sleep(0.3)
    for k in range(0, NUM, 5):
        l = int(lens[k])
        out = np.zeros((1, l), dtype=np.float32)
        w.pull(np.array([k], dtype=np.int64), out)
        assert np.allclose(out, float(world)), f"key {k} after intent: {out}"
    w.barrier()
    w.finalize()
    s.shutdown()


def test_non_uniform_lengths_ws2():
    run_dist(2, _non_uniform_dist, timeout=240)


def _four_channels(rank, world):
    import adapm_amd

    adapm_amd.setup(num_keys=64, num_threads=1, device="cpu", num_channels=4,
                    max_sync_per_sec=2000.0)
    s = adapm_amd.Server(4)
    w = adapm_amd.Worker(0, s)
    w.barrier()
    keys = np.arange(64, dtype=np.int64)
    w.push(keys, np.ones((64, 4), dtype=np.float32))
    w.barrier()
    out = np.zeros((64, 4), dtype=np.float32)
    w.pull(keys, out)
    assert np.allclose(out, float(world))
    w.intent(keys[:16], 1, 1_000_000)
    w.wait_sync()
    w.wait_sync()
    out2 = np.zeros((16, 4), dtype=np.float32)
    w.pull(keys[:16], out2)
    assert np.allclose(out2, float(world))
    w.barrier()
    w.finalize()
    s.shutdown()


def test_four_channels_ws3():
    run_dist(3, _four_channels, timeout=240)


def test_dmlc_env_fallback(monkeypatch):
    import adapm_amd
    from adapm_amd import runtime

    adapm_amd._SETUP.clear()
    runtime._RUNTIME = None
    monkeypatch.delenv("RANK", raising=False)
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    monkeypatch.setenv("DMLC_RANK", "0")
    monkeypatch.setenv("DMLC_NUM_SERVER", "1")
    rt = runtime.init_runtime(num_channels=1, device="cpu")
    assert rt.rank == 0 and rt.world == 1
    runtime._RUNTIME = None


def test_techniques_aliases():
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.setup(num_keys=4, num_threads=1, use_techniques="replication")
    assert adapm_amd._SETUP["techniques"] == adapm_amd.TECH_REPLICATION_ONLY
    adapm_amd.setup(num_keys=4, num_threads=1, use_techniques="ALL")
    assert adapm_amd._SETUP["techniques"] == adapm_amd.TECH_ALL
    adapm_amd._SETUP.clear()


def test_fused_step_guards_cpu():
    """kge_step_fused must refuse a CPU store (it is the single-rank GPU
    fast path; callers fall back to the classic pull/kernel/push path)."""
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=16, num_threads=1, device="cpu")
    s = adapm_amd.Server(8)
    k = torch.zeros(2, dtype=torch.int64)
    with pytest.raises(RuntimeError, match="GPU fast path"):
        s.raw.kge_step_fused(k, k, k, torch.zeros(4, dtype=torch.int64), 2, 4, 0.1, 1e-6)
    s.shutdown()
    adapm_amd._SETUP.clear()
