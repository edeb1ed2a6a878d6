"""Multi-process (gloo over loopback) basics: remote ops via sync rounds,
static ownership (no intents yet)."""
import torch

from dist_helper import run_dist


def _setup(rank, world, num_keys=64, lens=4, threads=1, **kw):
    import adapm_amd

    adapm_amd.setup(num_keys=num_keys, num_threads=threads, device="cpu",
                    max_sync_per_sec=500.0, **kw)
    s = adapm_amd.Server(lens)
    w = adapm_amd.Worker(0, s)
    return adapm_amd, s, w


def _remote_roundtrip(rank, world):
    _, s, w = _setup(rank, world)
    w.barrier()
    # every rank pushes 1.0 to every key (most keys are remote)
    keys = torch.arange(64)
    vals = torch.ones(64, 4)
    w.push(keys, vals)  # waits for acks
    w.barrier()
    out = torch.zeros(64, 4)
    w.pull(keys, out)  # remote pulls round-trip through sync rounds
    assert torch.equal(out, torch.full((64, 4), float(world))), out[:4]
    w.barrier()
    w.finalize()
    s.shutdown()


def test_remote_roundtrip_ws2():
    run_dist(2, _remote_roundtrip, timeout=180)


def test_remote_roundtrip_ws3():
    run_dist(3, _remote_roundtrip, timeout=180)


def _allreduce_and_barrier(rank, world):
    _, s, w = _setup(rank, world)
    total = w.allreduce(float(rank + 1))
    assert total == sum(r + 1 for r in range(world))
    t = w.allreduce(torch.tensor([1.0, 2.0]))
    assert torch.equal(t, torch.tensor([1.0 * world, 2.0 * world]))
    w.barrier()
    w.finalize()
    s.shutdown()


def test_allreduce_ws2():
    run_dist(2, _allreduce_and_barrier, timeout=180)


def _set_remote(rank, world):
    _, s, w = _setup(rank, world)
    w.barrier()
    if rank == 1:
        w.set(torch.tensor([0]), torch.tensor([[9.0, 9.0, 9.0, 9.0]]))  # key 0 lives on rank 0
    w.barrier()
    out = torch.zeros(1, 4)
    w.pull(torch.tensor([0]), out)
    assert torch.equal(out, torch.full((1, 4), 9.0))
    w.barrier()
    w.finalize()
    s.shutdown()


def test_set_remote_ws2():
    run_dist(2, _set_remote, timeout=180)
