"""Sampling scheme matrix (mirrors reference tests/test_sampling.cc +
run_tests.sh variants): each scheme returns valid keys with correct
values; with/without replacement; keys relocated away first to stress
non-local paths."""
import numpy as np
import torch

from dist_helper import run_dist

NUM_KEYS = 64
LEN = 4


def _sampling(rank, world, scheme, with_replacement):
    import adapm_amd

    adapm_amd.setup(num_keys=NUM_KEYS, num_threads=1, device="cpu", max_sync_per_sec=2000.0)
    s = adapm_amd.Server(LEN)
    s.enable_sampling_support(scheme, with_replacement, "uniform", 0, NUM_KEYS)
    w = adapm_amd.Worker(0, s)
    w.barrier()

    # give every key a recognizable value: val[:] = key
    if rank == 0:
        keys = np.arange(NUM_KEYS, dtype=np.int64)
        vals = np.tile(np.arange(NUM_KEYS, dtype=np.float32)[:, None], (1, LEN))
        w.set(keys, vals)
    w.wait_sync()
    w.barrier()

    # stress remote paths: relocate some keys away from their home
    if world > 1 and rank == 0:
        w.intent(np.arange(1, NUM_KEYS, 7, dtype=np.int64), 1, 1_000_000)
        import time

        time.sleep(0.2)
    w.barrier()

    for trial in range(10):
        K = 8
        sid = w.prepare_sample(K, w.current_clock(), w.current_clock() + 50)
        keys = np.zeros(K, dtype=np.int64)
        vals = np.zeros((K, LEN), dtype=np.float32)
        w.pull_sample(sid, keys, vals)
        assert ((keys >= 0) & (keys < NUM_KEYS)).all(), keys
        for i, k in enumerate(keys):
            assert np.allclose(vals[i], float(k)), \
                f"rank {rank} scheme {scheme}: key {k} has vals {vals[i]}"
        if not with_replacement:
            assert len(set(keys.tolist())) == K, f"duplicates in WOR sample: {keys}"
        w.finish_sample(sid)
        w.advance_clock()

    # partial consumption: pull a K-sample in two halves
    sid = w.prepare_sample(8, w.current_clock(), w.current_clock() + 50)
    k1 = np.zeros(4, dtype=np.int64)
    v1 = np.zeros((4, LEN), dtype=np.float32)
    w.pull_sample(sid, k1, v1)
    k2 = np.zeros(4, dtype=np.int64)
    v2 = np.zeros((4, LEN), dtype=np.float32)
    w.pull_sample(sid, k2, v2)
    w.finish_sample(sid)

    w.barrier()
    w.finalize()
    s.shutdown()


def test_sampling_naive_ws2():
    run_dist(2, _sampling, "naive", True, timeout=240)


def test_sampling_preloc_ws2():
    run_dist(2, _sampling, "preloc", True, timeout=240)


def test_sampling_pool_ws2():
    run_dist(2, _sampling, "pool", True, timeout=240)


def test_sampling_local_ws2():
    run_dist(2, _sampling, "local", True, timeout=240)


def test_sampling_naive_wor_ws2():
    run_dist(2, _sampling, "naive", False, timeout=240)


def test_sampling_local_wor_ws2():
    run_dist(2, _sampling, "local", False, timeout=240)


def test_sampling_single_rank_local():
    _sampling(0, 1, "local", True)


def test_log_uniform_distribution():
    from adapm_amd.sampling import LogUniform

    d = LogUniform(0, 1000, seed=7)
    ks = d.draw(20000)
    assert (ks >= 0).all() and (ks < 1000).all()
    # log-uniform: small keys much more frequent
    assert (ks < 100).mean() > (ks >= 900).mean() * 3


def test_unigram_distribution():
    from adapm_amd.sampling import Unigram

    counts = np.array([100, 10, 1, 1])
    d = Unigram(counts, None, 0.75, seed=3)
    ks = d.draw(20000)
    freq = np.bincount(ks, minlength=4) / 20000
    assert freq[0] > freq[1] > freq[2] * 2


def test_alias_table_distribution():
    """Alias-table draws match the target unigram^power distribution."""
    from adapm_amd.sampling import Unigram

    counts = np.array([100, 50, 10, 5, 1], dtype=np.float64)
    d = Unigram(counts, None, 1.0, seed=9)
    ks = d.draw(200000)
    freq = np.bincount(ks, minlength=5) / 200000
    target = counts / counts.sum()
    assert np.allclose(freq, target, atol=0.01), (freq, target)
