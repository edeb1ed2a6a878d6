"""Host-spill arena: the store keeps working when values exceed the
device-arena budget (BASELINE config 5 capability: table > HBM)."""
import numpy as np
import pytest
import torch


def _run_spill(device):
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    # 1000 keys x 256 floats = ~1 MB of values, device arena capped at
    # ~0.25 MB -> most keys spill to the host arena
    adapm_amd.setup(num_keys=1000, num_threads=1, device=device,
                    device_cap_gb=0.25e-3, host_spill_gb=0.01)
    s = adapm_amd.Server(256)
    st = s.stats()
    assert st["host_spill_in_use"] > 0, st
    w = adapm_amd.Worker(0, s)
    g = torch.Generator().manual_seed(0)
    keys = torch.randperm(1000, generator=g)[:300].to(torch.int64)
    vals = torch.randn(300, 256, generator=g)
    if device.startswith("cuda"):
        vals = vals.cuda()
    w.push(keys, vals)
    out = torch.zeros_like(vals)
    w.pull(keys, out)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    assert torch.equal(out, vals)
    w.push(keys, vals)
    w.pull(keys, out)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    assert torch.allclose(out, 2 * vals)
    s.shutdown()


def test_host_spill_cpu():
    _run_spill("cpu")


@pytest.mark.gpu
def test_host_spill_gpu():
    _run_spill("cuda:0")


def _mk_tiered(device):
    """400 keys x 64 floats; device arena sized for ~100 rows, rest spill."""
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=400, num_threads=1, device=device,
                    device_cap_gb=100 * 64 * 4 / 2**30, host_spill_gb=0.01)
    s = adapm_amd.Server(64)
    w = adapm_amd.Worker(0, s)
    return adapm_amd, s, w


def _run_rebalance(device):
    import torch as T

    adapm_amd, s, w = _mk_tiered(device)
    keys = np.arange(400, dtype=np.int64)
    vals = torch.arange(400, dtype=torch.float32)[:, None].expand(400, 64).contiguous()
    if device.startswith("cuda"):
        vals = vals.cuda()
    w.set(keys, vals)
    tiers = np.array([s.raw.key_tier(int(k)) for k in keys])
    assert (tiers == 0).sum() == 100 and (tiers == 1).sum() == 300, tiers

    # hammer 20 spilled keys; touch device keys once so they have heat=1
    hot = keys[tiers == 1][:20]
    out = torch.zeros(20, 64, device=vals.device)
    for _ in range(30):
        w.pull(hot, out)
    moved = s.raw.rebalance_spill(64)
    assert moved == 20, moved
    for k in hot:
        assert s.raw.key_tier(int(k)) == 0, k
    # total device-resident count unchanged (swaps, not growth)
    tiers2 = np.array([s.raw.key_tier(int(k)) for k in keys])
    assert (tiers2 == 0).sum() == 100, tiers2.sum()

    # ALL values must be intact after the swap
    out_all = torch.zeros(400, 64, device=vals.device)
    w.pull(keys, out_all)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    assert torch.equal(out_all.cpu(), vals.cpu())

    # pushes after the swap land in the new slots
    w.push(hot, torch.ones(20, 64, device=vals.device))
    w.pull(hot, out)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    assert torch.equal(out.cpu(), vals.cpu()[hot] + 1)

    # a second rebalance with no new skew does nothing (hysteresis)
    assert s.raw.rebalance_spill(64) == 0
    assert s.stats()["spill_rebalance_moves"] == 20
    s.shutdown()


def test_spill_rebalance_cpu():
    _run_rebalance("cpu")


@pytest.mark.gpu
def test_spill_rebalance_gpu():
    _run_rebalance("cuda:0")


def test_rebalance_concurrent_with_workers_cpu():
    """The stop-the-world gate: worker threads hammer pulls/pushes while
    the main thread rebalances repeatedly; no op may observe a torn
    migration (values stay exact)."""
    import threading

    adapm_amd, s, w = _mk_tiered("cpu")
    keys = np.arange(400, dtype=np.int64)
    w.set(keys, torch.zeros(400, 64))
    pushes = np.zeros(400)
    stop = threading.Event()
    errs = []

    def hammer(seed):
        rng = np.random.default_rng(seed)
        local = np.zeros(400)
        try:
            while not stop.is_set():
                ks = rng.choice(400, size=32, replace=False).astype(np.int64)
                w.push(ks, np.ones((32, 64), dtype=np.float32))
                local[ks] += 1
                out = np.zeros((32, 64), dtype=np.float32)
                w.pull(ks, out)
        except Exception as e:  # pragma: no cover
            errs.append(e)
        results.append(local)

    results = []
    threads = [threading.Thread(target=hammer, args=(i,)) for i in range(2)]
    for t in threads:
        t.start()
    for _ in range(30):
        s.raw.rebalance_spill(256)
    stop.set()
    for t in threads:
        t.join(timeout=60)
    assert not errs, errs
    total = sum(results)
    out = np.zeros((400, 64), dtype=np.float32)
    w.pull(keys, out)
    assert np.allclose(out, total[:, None]), \
        f"mismatch at {np.where(np.abs(out[:, 0] - total) > 1e-3)[0]}"
    s.shutdown()


def _spill_dist_worker(rank, world):
    """World>1 with a spilled store: the sync protocol's extract /
    relocation-gather / refresh paths must work on host-arena rows
    (BASELINE config 5 is 8 GPUs + host spill). Exact-sum under churn."""
    import adapm_amd

    # 64 keys x 64 floats; device arena capped so most rows spill
    adapm_amd.setup(num_keys=64, num_threads=1, device="cpu",
                    device_cap_gb=16 * 64 * 4 / 2**30, host_spill_gb=0.01,
                    max_sync_per_sec=4000.0)
    s = adapm_amd.Server(64)
    assert s.stats()["host_spill_in_use"] > 0
    w = adapm_amd.Worker(0, s)
    w.barrier()
    rng = np.random.default_rng(rank)
    pushes = np.zeros(64)
    for i in range(120):
        keys = rng.choice(64, size=3, replace=False).astype(np.int64)
        if rng.random() < 0.4:
            w.intent(keys, w.current_clock() + 1,
                     w.current_clock() + int(rng.integers(2, 10)))
        w.push(keys, np.ones((3, 64), dtype=np.float32), async_=True)
        pushes[keys] += 1
        if rng.random() < 0.2:
            w.pull(keys, np.zeros((3, 64), dtype=np.float32))
        w.advance_clock()
    w.waitall()
    w.barrier()
    w.wait_sync(strong=True)
    w.barrier()
    total = w.allreduce(torch.tensor(pushes, dtype=torch.float32)).numpy()
    out = np.zeros((64, 64), dtype=np.float32)
    w.pull(np.arange(64, dtype=np.int64), out)
    assert np.allclose(out[:, 0], total, atol=1e-2), \
        f"rank {rank} mismatch at {np.where(np.abs(out[:, 0] - total) > 1e-2)[0]}"
    st = s.stats()
    assert st["relocations_out"] + st["replications"] > 0  # churn really happened
    w.barrier()
    w.finalize()
    s.shutdown()


def test_host_spill_dist_ws2():
    from dist_helper import run_dist

    run_dist(2, _spill_dist_worker, timeout=240)


@pytest.mark.gpu
def test_host_spill_dist_ws2_gpu():
    """Same protocol churn on a spilled GPU store (pinned host arena,
    device-visible zero-copy; 2 ranks share cuda:0 over gloo)."""
    import os

    from dist_helper import run_dist

    os.environ["ADAPM_FORCE_GLOO"] = "1"
    run_dist(2, _spill_dist_worker_gpu, timeout=240)


def _spill_dist_worker_gpu(rank, world):
    import os

    os.environ["ADAPM_FORCE_GLOO"] = "1"
    _spill_dist_worker_impl(rank, world, "cuda:0")


def _spill_dist_worker_impl(rank, world, device):
    import adapm_amd

    adapm_amd.setup(num_keys=64, num_threads=1, device=device,
                    device_cap_gb=16 * 64 * 4 / 2**30, host_spill_gb=0.01,
                    max_sync_per_sec=4000.0)
    s = adapm_amd.Server(64)
    assert s.stats()["host_spill_in_use"] > 0
    w = adapm_amd.Worker(0, s)
    w.barrier()
    rng = np.random.default_rng(rank)
    pushes = np.zeros(64)
    for i in range(120):
        keys = rng.choice(64, size=3, replace=False).astype(np.int64)
        if rng.random() < 0.4:
            w.intent(keys, w.current_clock() + 1,
                     w.current_clock() + int(rng.integers(2, 10)))
        w.push(keys, np.ones((3, 64), dtype=np.float32), async_=True)
        pushes[keys] += 1
        if rng.random() < 0.2:
            w.pull(keys, np.zeros((3, 64), dtype=np.float32))
        w.advance_clock()
    w.waitall()
    w.barrier()
    w.wait_sync(strong=True)
    w.barrier()
    total = w.allreduce(torch.tensor(pushes, dtype=torch.float32)).numpy()
    out = np.zeros((64, 64), dtype=np.float32)
    w.pull(np.arange(64, dtype=np.int64), out)
    assert np.allclose(out[:, 0], total, atol=1e-2), \
        f"rank {rank} mismatch at {np.where(np.abs(out[:, 0] - total) > 1e-2)[0]}"
    w.barrier()
    w.finalize()
    s.shutdown()
