"""End-to-end app smoke tests (reference tests/run_apps.sh): word2vec,
MF (all 3 schedules), CTR, simple — single rank CPU + distributed CPU +
GPU twins where it matters."""
import os

import numpy as np
import pytest
import torch

from dist_helper import run_dist


def _fresh(device="cpu", **kw):
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    return adapm_amd


def test_simple_single():
    adapm = _fresh()
    from adapm_amd.models.simple import run_simple

    stats = run_simple(iterations=20, num_keys=100, device="cpu")
    assert stats["pull_keys"] >= 60


def _simple_dist(rank, world):
    from adapm_amd.models.simple import run_simple

    run_simple(iterations=30, num_keys=200, device="cpu")


def test_simple_ws3():
    run_dist(3, _simple_dist, timeout=180)


def _w2v(device):
    adapm = _fresh()
    from adapm_amd.models.word2vec import (W2VConfig, Word2Vec,
                                           make_synthetic_sentences, syn0)

    V = 500
    cfg = W2VConfig(vocab_size=V, dim=32, negative=3, window=3, lr=0.05)
    adapm.setup(num_keys=cfg.num_keys, num_threads=1, device=device)
    server = adapm.Server(cfg.row)
    counts = np.arange(V, 0, -1)  # fake frequency counts
    server.enable_sampling_support("local", True, "unigram", 0, V, counts=counts)
    # sampling draws WORD ids; the model maps them to syn1 keys itself
    worker = adapm.Worker(0, server)
    model = Word2Vec(cfg, server, worker)
    model.set_vocab_counts(counts)
    model.init_embeddings()
    sents = make_synthetic_sentences(100, V, seed=2)
    epoch_losses = []
    for epoch in range(4):
        ctr, ctx = model.pairs_from_sentences(sents)
        ls = [model.train_pairs(ctr[i:i + 2048], ctx[i:i + 2048], sync_loss=True)
              for i in range(0, len(ctr), 2048)]
        epoch_losses.append(float(np.mean(ls)))
    model.drain()
    assert epoch_losses[-1] < epoch_losses[0], f"no learning {epoch_losses}"
    path = "/tmp/w2v_test.txt"
    model.export_text(path, max_words=10)
    lines = open(path).read().splitlines()
    assert lines[0] == "10 32" and len(lines) == 11
    os.remove(path)
    worker.finalize()
    server.shutdown()


def test_w2v_cpu():
    _w2v("cpu")


@pytest.mark.gpu
def test_w2v_gpu():
    _w2v("cuda:0")


def _mf(device, schedule):
    adapm = _fresh()
    from adapm_amd.models.mf import MF, MFConfig, make_synthetic_ratings

    cfg = MFConfig(num_rows=300, num_cols=100, rank=16, batch_nnz=1024, lr=0.05)
    adapm.setup(num_keys=cfg.num_keys, num_threads=1, device=device)
    server = adapm.Server(cfg.row)
    worker = adapm.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()
    rows, cols, ratings = make_synthetic_ratings(5000, cfg.num_rows, cfg.num_cols, seed=3)
    ep = getattr(model, f"epoch_{schedule}")
    losses = [ep(rows, cols, ratings) for _ in range(4)]
    assert losses[-1] < losses[0], f"{schedule}: {losses}"
    tl = model.test_loss(rows[:500], cols[:500], ratings[:500])
    assert np.isfinite(tl)
    worker.finalize()
    server.shutdown()


def test_mf_plain_sgd_cpu():
    _mf("cpu", "plain_sgd")


def test_mf_dsgd_cpu():
    _mf("cpu", "dsgd")


def test_mf_columnwise_cpu():
    _mf("cpu", "columnwise")


@pytest.mark.gpu
def test_mf_plain_sgd_gpu():
    _mf("cuda:0", "plain_sgd")


def _mf_dist(rank, world, schedule):
    import adapm_amd
    from adapm_amd.models.mf import MF, MFConfig, make_synthetic_ratings

    cfg = MFConfig(num_rows=200, num_cols=80, rank=8, batch_nnz=512, lr=0.05)
    adapm_amd.setup(num_keys=cfg.num_keys, num_threads=1, device="cpu",
                    max_sync_per_sec=2000.0)
    server = adapm_amd.Server(cfg.row)
    worker = adapm_amd.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()
    rows, cols, ratings = make_synthetic_ratings(3000, cfg.num_rows, cfg.num_cols, seed=4)
    # each rank trains on a row partition (reference data partitioning)
    mine = rows % world == rank
    ep = getattr(model, f"epoch_{schedule}")
    losses = [ep(rows[mine], cols[mine], ratings[mine]) for _ in range(3)]
    total_first = worker.allreduce(losses[0])
    total_last = worker.allreduce(losses[-1])
    assert total_last < total_first
    worker.barrier()
    worker.finalize()
    server.shutdown()


def test_mf_dsgd_ws2():
    run_dist(2, _mf_dist, "dsgd", timeout=300)


def test_mf_columnwise_ws2():
    run_dist(2, _mf_dist, "columnwise", timeout=300)


def _ctr(device):
    adapm = _fresh()
    from adapm_amd.models.ctr import CTRConfig, WideAndDeep, make_synthetic_ctr

    cfg = CTRConfig(num_features=2000, dim=16, fields=4, hidden=32, batch_size=512)
    adapm.setup(num_keys=cfg.num_features, num_threads=1, device=device)
    server = adapm.Server(cfg.row)
    worker = adapm.Worker(0, server)
    model = WideAndDeep(cfg, server, worker)
    model.init_embeddings()
    feats, labels = make_synthetic_ctr(4096, cfg.num_features, cfg.fields, seed=6)
    losses = []
    for epoch in range(6):
        for i in range(0, len(feats), cfg.batch_size):
            losses.append(model.train_batch(feats[i:i + cfg.batch_size],
                                            labels[i:i + cfg.batch_size]))
    model.drain()
    assert losses[-1] < losses[0] * 0.98, f"no learning: {losses[0]} -> {losses[-1]}"
    worker.finalize()
    server.shutdown()


def test_ctr_cpu():
    _ctr("cpu")


@pytest.mark.gpu
def test_ctr_gpu():
    _ctr("cuda:0")


def test_mf_checkpoint_roundtrip(tmp_path):
    adapm = _fresh()
    from adapm_amd.models.mf import (MF, MFConfig, load_factors,
                                     make_synthetic_ratings, save_factors)

    cfg = MFConfig(num_rows=100, num_cols=50, rank=8)
    adapm.setup(num_keys=cfg.num_keys, num_threads=1, device="cpu")
    server = adapm.Server(cfg.row)
    worker = adapm.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()
    path = str(tmp_path / "mf.npz")
    save_factors(model, path)
    before = np.zeros((5, cfg.row), dtype=np.float32)
    worker.pull(np.arange(5, dtype=np.int64), before)
    worker.set(np.arange(5, dtype=np.int64), np.zeros((5, cfg.row), dtype=np.float32))
    load_factors(model, path)
    after = np.zeros((5, cfg.row), dtype=np.float32)
    worker.pull(np.arange(5, dtype=np.int64), after)
    assert np.allclose(before, after)
    worker.finalize()
    server.shutdown()
