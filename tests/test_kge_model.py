"""End-to-end KGE ComplEx model on the store (CPU tier + GPU twin):
training reduces loss; eval and checkpoint round-trip work."""
import os

import numpy as np
import pytest
import torch


def _run_kge(device):
    import adapm_amd
    from adapm_amd.models.kge import ComplEx, ComplExConfig, make_synthetic_triples

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    E, R = 500, 20
    adapm_amd.setup(num_keys=E + R, num_threads=1, device=device)
    server = adapm_amd.Server(2 * 64)
    server.enable_sampling_support("local", True, "uniform", 0, E)
    worker = adapm_amd.Worker(0, server)
    cfg = ComplExConfig(num_entities=E, num_relations=R, dim=64, neg_samples=4,
                        batch_size=128, lr=0.2)
    model = ComplEx(cfg, server, worker)
    model.init_embeddings()

    triples = make_synthetic_triples(512, E, R, seed=1)
    losses = []
    for epoch in range(8):
        for i in range(0, len(triples), cfg.batch_size):
            losses.append(model.train_batch(triples[i:i + cfg.batch_size]))
    model.drain()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    assert losses[-1] < losses[0] * 0.8, f"no learning: {losses[0]} -> {losses[-1]}"

    ev = model.evaluate(triples[:64], num_candidates=100)
    assert 0.0 < ev["mrr"] <= 1.0 and ev["hits@10"] >= 0.0

    # filtered eval (reference computes filtered + raw ranks,
    # knowledge_graph_embeddings.cc:544-712): excluding known-true
    # competitors can only improve (or tie) every rank
    evf = model.evaluate_full(triples[:48], filter_triples=triples)
    assert evf["mrr"] >= evf["mrr_raw"] - 1e-9
    assert evf["mr"] <= evf["mr_raw"] + 1e-9
    assert 0.0 < evf["mrr"] <= 1.0
    # a (s, r) with two true objects: the other true object must not
    # count against the evaluated one. Build a duplicate-(s,r) case:
    dup = np.array([[1, 0, 2], [1, 0, 3]], dtype=np.int64)
    e1 = model.evaluate_full(dup, filter_triples=dup)
    e2 = model.evaluate_full(dup)  # raw only
    assert e1["mrr"] >= e2["mrr"] - 1e-9

    # checkpoint round-trip
    path = "/tmp/kge_ckpt_test.npz"
    model.save_checkpoint(path)
    before = np.zeros((4, cfg.row), dtype=np.float32)
    worker.pull(np.arange(4, dtype=np.int64), before)
    # clobber and restore
    worker.set(np.arange(4, dtype=np.int64), np.zeros((4, cfg.row), dtype=np.float32))
    model.load_checkpoint(path)
    after = np.zeros((4, cfg.row), dtype=np.float32)
    worker.pull(np.arange(4, dtype=np.int64), after)
    assert np.allclose(before, after, atol=1e-6)
    os.remove(path)

    worker.finalize()
    server.shutdown()


def test_kge_model_cpu():
    _run_kge("cpu")


@pytest.mark.gpu
def test_kge_model_gpu():
    _run_kge("cuda:0")


def _kge_dist(rank, world):
    import adapm_amd
    from adapm_amd.models.kge import ComplEx, ComplExConfig, make_synthetic_triples

    E, R = 300, 10
    adapm_amd.setup(num_keys=E + R, num_threads=1, device="cpu", max_sync_per_sec=2000.0)
    server = adapm_amd.Server(2 * 32)
    server.enable_sampling_support("local", True, "uniform", 0, E)
    worker = adapm_amd.Worker(0, server)
    cfg = ComplExConfig(num_entities=E, num_relations=R, dim=32, neg_samples=2,
                        batch_size=64, lr=0.2, lookahead=2)
    model = ComplEx(cfg, server, worker)
    model.init_embeddings()
    triples = make_synthetic_triples(256, E, R, seed=rank)
    first = last = None
    for epoch in range(4):
        for i in range(0, len(triples), cfg.batch_size):
            b = triples[i:i + cfg.batch_size]
            model.signal_intent(b, worker.current_clock() + 1, worker.current_clock() + 3)
            loss = model.train_batch(b)
            first = loss if first is None else first
            last = loss
            worker.advance_clock()
    model.drain()
    total_first = worker.allreduce(first)
    total_last = worker.allreduce(last)
    assert total_last < total_first, f"no learning: {total_first} -> {total_last}"
    ev = model.evaluate(triples[:32], num_candidates=64)
    assert ev["n"] == 32 * world
    worker.barrier()
    worker.finalize()
    server.shutdown()


def test_kge_model_distributed_ws2():
    from dist_helper import run_dist

    run_dist(2, _kge_dist, timeout=300)
