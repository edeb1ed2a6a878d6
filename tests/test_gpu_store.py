"""GPU (MI355X) tests of the HBM store: exact semantics of the HIP
gather/scatter/extract/refresh kernels through the Server API."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def make_server(num_keys=1000, lens=64, **kw):
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=num_keys, num_threads=1, device="cuda:0", **kw)
    return adapm_amd, adapm_amd.Server(lens)


def test_gpu_push_pull_exact():
    adapm, s = make_server()
    w = adapm.Worker(0, s)
    g = torch.Generator().manual_seed(0)
    keys = torch.randperm(1000, generator=g)[:128].to(torch.int64)
    vals = torch.randn(128, 64, generator=g).cuda()
    assert w.push(keys, vals) == -1
    out = torch.zeros(128, 64, device="cuda")
    assert w.pull(keys, out) == -1
    torch.cuda.synchronize()
    assert torch.equal(out, vals)
    # additive
    w.push(keys, vals)
    w.pull(keys, out)
    torch.cuda.synchronize()
    assert torch.allclose(out, 2 * vals)
    # set overwrites
    w.set(keys, vals)
    w.pull(keys, out)
    torch.cuda.synchronize()
    assert torch.equal(out, vals)
    s.shutdown()


def test_gpu_repeated_keys_atomic_accumulate():
    adapm, s = make_server(num_keys=16, lens=32)
    w = adapm.Worker(0, s)
    keys = torch.zeros(512, dtype=torch.int64)  # all the same key
    vals = torch.ones(512, 32, device="cuda")
    w.push(keys, vals)
    out = torch.zeros(1, 32, device="cuda")
    w.pull(torch.tensor([0]), out)
    torch.cuda.synchronize()
    assert torch.allclose(out, torch.full((1, 32), 512.0, device="cuda"))
    s.shutdown()


def test_gpu_cpu_tensor_interop():
    adapm, s = make_server()
    w = adapm.Worker(0, s)
    keys = np.array([3, 7], dtype=np.int64)
    vals = np.random.default_rng(1).normal(size=(2, 64)).astype(np.float32)
    w.push(keys, vals)
    out = np.zeros((2, 64), dtype=np.float32)
    w.pull(keys, out)
    assert np.allclose(out, vals, atol=1e-6)
    s.shutdown()


def test_gpu_odd_length_rows():
    """Non-multiple-of-4 lengths exercise the scalar tail path."""
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    adapm_amd.setup(num_keys=10, num_threads=1, device="cuda:0")
    lens = torch.tensor([3, 5, 7, 3, 5, 7, 3, 5, 7, 3])
    s = adapm_amd.Server(lens)
    w = adapm_amd.Worker(0, s)
    keys = torch.tensor([0, 1, 2])
    vals = torch.arange(15, dtype=torch.float32).cuda()  # 3+5+7
    w.push(keys, vals)
    out = torch.zeros(15, device="cuda")
    w.pull(keys, out)
    torch.cuda.synchronize()
    assert torch.equal(out, vals)
    s.shutdown()


def test_gpu_native_extension_loaded():
    """Fail loudly if the native path is not the one running on GPU."""
    from adapm_amd import _C

    assert _C.hip_available(), "HIP reports no device on a GPU box"
    import adapm_amd

    so = _C.__file__
    assert so.startswith("/root/") or "adapm_amd" in so, so
