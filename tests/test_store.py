"""Single-rank store semantics (CPU backend of the same native core)."""
import numpy as np
import pytest
import torch

import adapm_amd


def make_server(num_keys=100, lens=8, threads=1, **kw):
    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=num_keys, num_threads=threads, device="cpu", **kw)
    return adapm_amd.Server(lens)


def test_push_pull_set_roundtrip():
    s = make_server()
    w = adapm_amd.Worker(0, s)
    keys = torch.tensor([3, 10, 42])
    vals = torch.arange(24, dtype=torch.float32).reshape(3, 8)
    assert w.push(keys, vals) == -1
    out = torch.zeros(3, 8)
    assert w.pull(keys, out) == -1
    assert torch.equal(out, vals)
    w.push(keys, vals)
    w.pull(keys, out)
    assert torch.equal(out, 2 * vals)
    w.set(keys, vals)
    w.pull(keys, out)
    assert torch.equal(out, vals)
    s.shutdown()


def test_numpy_overloads():
    s = make_server()
    w = adapm_amd.Worker(0, s)
    keys = np.array([1, 2], dtype=np.int64)
    vals = np.ones((2, 8), dtype=np.float32)
    w.push(keys, vals)
    out = np.zeros((2, 8), dtype=np.float32)
    w.pull(keys, out)
    assert np.array_equal(out, vals)
    s.shutdown()


def test_non_uniform_lengths():
    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=10, num_threads=1, device="cpu")
    lens = torch.tensor([2, 4, 8, 2, 4, 8, 2, 4, 8, 2])
    s = adapm_amd.Server(lens)
    w = adapm_amd.Worker(0, s)
    assert w.get_key_size(0) == 2 and w.get_key_size(2) == 8
    keys = torch.tensor([0, 1, 2])
    vals = torch.arange(14, dtype=torch.float32)  # 2+4+8
    w.push(keys, vals)
    out = torch.zeros(14)
    w.pull(keys, out)
    assert torch.equal(out, vals)
    # single-key pulls see the right slices
    o0 = torch.zeros(2)
    w.pull(torch.tensor([0]), o0)
    assert torch.equal(o0, vals[:2])
    o2 = torch.zeros(8)
    w.pull(torch.tensor([2]), o2)
    assert torch.equal(o2, vals[6:14])
    s.shutdown()


def test_repeated_keys_accumulate():
    s = make_server()
    w = adapm_amd.Worker(0, s)
    keys = torch.tensor([7, 7, 7])
    vals = torch.ones(3, 8)
    w.push(keys, vals)
    out = torch.zeros(1, 8)
    w.pull(torch.tensor([7]), out)
    assert torch.equal(out, torch.full((1, 8), 3.0))
    s.shutdown()


def test_pull_if_local_single_rank():
    s = make_server()
    w = adapm_amd.Worker(0, s)
    out = torch.zeros(1, 8)
    assert w.pull_if_local(torch.tensor([5]), out)  # single rank: everything local
    assert w.is_local(5)
    s.shutdown()


def test_wrong_val_length_raises():
    s = make_server()
    w = adapm_amd.Worker(0, s)
    with pytest.raises(ValueError):
        w.push(torch.tensor([1]), torch.zeros(3))
    s.shutdown()


def test_exact_sum_async_single_rank():
    """Mirror of reference test_dynamic_allocation invariant on 1 rank:
    many async pushes of {1,2} to one key sum exactly."""
    s = make_server(num_keys=4, lens=2)
    w = adapm_amd.Worker(0, s)
    runs = 2000
    key = torch.tensor([1])
    v = torch.tensor([[1.0, 2.0]])
    for _ in range(runs):
        w.push(key, v, async_=True)
    w.waitall()
    out = torch.zeros(1, 2)
    w.pull(key, out)
    assert out[0, 0].item() == runs and out[0, 1].item() == 2 * runs
    s.shutdown()


def test_staggered_push():
    s = make_server(num_keys=1000, lens=4)
    w = adapm_amd.Worker(0, s)
    keys = torch.arange(1000)
    vals = torch.ones(1000, 4)
    w.staggered_push(keys, vals, chunk=128)
    out = torch.zeros(1000, 4)
    w.pull(keys, out)
    assert torch.equal(out, vals)
    s.shutdown()


def test_out_of_range_key_raises():
    s = make_server(num_keys=10, lens=4)
    w = adapm_amd.Worker(0, s)
    with pytest.raises(RuntimeError):
        w.push(torch.tensor([11]), torch.zeros(1, 4))
    with pytest.raises(RuntimeError):
        w.pull(torch.tensor([-1]), torch.zeros(1, 4))
    with pytest.raises(RuntimeError):
        w.intent(torch.tensor([99]), 1, 5)
    s.shutdown()
