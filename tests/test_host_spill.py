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
