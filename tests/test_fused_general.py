"""World>1 fused slab-direct path (kge_step_fused_general): samples whose
keys are all local run the fused offsets-mode kernel (including replica
rows created by intent), the rest take the classic pull/kernel/push
path, and the results must match the classic math (unique keys, so
in-place fused updates == pulled-snapshot deltas). VERDICT r01 item 3."""
import numpy as np
import pytest
import torch

from dist_helper import run_dist

ENT = 32
REL = 4
DIM = 8
NEG = 2


def _expected_after_classic(init_rows, s, r, o, negs, lr, eps):
    """Classic-path reference: deltas from the CPU step kernel applied to
    the pulled snapshot (valid when all keys in the batch are unique)."""
    import adapm_amd
    from adapm_amd import _C

    B = len(s)
    row = 2 * DIM
    sv = torch.from_numpy(np.stack([init_rows[k] for k in s])).contiguous()
    rv = torch.from_numpy(np.stack([init_rows[k] for k in r])).contiguous()
    ov = torch.from_numpy(np.stack([init_rows[k] for k in o])).contiguous()
    nv = torch.from_numpy(np.stack([init_rows[k] for k in negs])).contiguous()
    ds, dr, do, dn = (torch.empty_like(t) for t in (sv, rv, ov, nv))
    loss = torch.empty(B, dtype=torch.float32)
    _C.kge_complex_step(sv, rv, ov, nv, ds, dr, do, dn, loss, NEG, DIM, lr, eps)
    exp = {int(k): init_rows[k].copy() for k in np.concatenate([s, r, o, negs])}
    for i, k in enumerate(s):
        exp[int(k)] += ds[i].numpy()
    for i, k in enumerate(r):
        exp[int(k)] += dr[i].numpy()
    for i, k in enumerate(o):
        exp[int(k)] += do[i].numpy()
    for i, k in enumerate(negs):
        exp[int(k)] += dn[i].numpy()
    return exp, loss


def _fused_general_worker(rank, world, use_intent, device="cpu"):
    import os

    if device != "cpu":
        os.environ["ADAPM_FORCE_GLOO"] = "1"  # 2 ranks share one GPU
    import adapm_amd
    from adapm_amd.models.kge import ComplEx, ComplExConfig

    adapm_amd.setup(num_keys=ENT + REL, num_threads=1, device=device,
                    max_sync_per_sec=4000.0)
    server = adapm_amd.Server(2 * DIM)
    worker = adapm_amd.Worker(0, server)
    cfg = ComplExConfig(num_entities=ENT, num_relations=REL, dim=DIM,
                        neg_samples=NEG, batch_size=4, seed=3)
    model = ComplEx(cfg, server, worker)
    worker.barrier()

    # known initial values for every key (deterministic across ranks)
    rng = np.random.default_rng(42)
    init_rows = rng.standard_normal((ENT + REL, 2 * DIM)).astype(np.float32) * 0.1
    init_rows[:, DIM:] = np.abs(init_rows[:, DIM:])  # AdaGrad accums >= 0
    if rank == 0:
        worker.set(np.arange(ENT + REL, dtype=np.int64), init_rows)
    worker.wait_sync(strong=True)
    worker.barrier()

    # unique keys: distinct s/o/negs, one relation
    triples = np.array([[0, 1, 5], [2, 0, 9], [4, 2, 13], [6, 3, 17]], dtype=np.int64)
    s, r, o = model.keys_of(triples)
    negs = np.array([20, 21, 22, 23, 24, 25, 26, 27], dtype=np.int64)  # 4*NEG unique

    if use_intent and rank == 0:
        all_keys = np.concatenate([s, r, o, negs])
        worker.intent(all_keys, worker.current_clock(), worker.current_clock() + 50)
        # wait until most keys are locally present (replicated/relocated)
        import time

        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            n_local = sum(worker.is_local(int(k)) for k in all_keys)
            if n_local == len(all_keys):
                break
            time.sleep(0.05)

    if rank == 0:
        model.rng = np.random.default_rng(7)  # unused (negs passed implicitly)
        # run the general fused step with our fixed negatives by calling
        # the raw API (the model wrapper draws negatives itself)
        loss, missed = server.raw.kge_step_fused_general(
            torch.from_numpy(s), torch.from_numpy(r), torch.from_numpy(o),
            torch.from_numpy(negs), NEG, DIM, cfg.lr, cfg.eps)
        if use_intent:
            assert missed.numel() == 0, f"all keys local but {missed.numel()} samples missed"
        if missed.numel():
            midx = missed.numpy()
            sub_negs = negs.reshape(len(triples), NEG)[midx].reshape(-1)
            mloss = model.train_batch(triples[midx], sync_loss=False, neg_keys=sub_negs)
            loss = torch.cat([loss, mloss])
        assert loss.numel() == len(triples)
        assert bool(torch.isfinite(loss).all())
        model.drain()

    worker.waitall()
    worker.barrier()
    worker.wait_sync(strong=True)
    worker.barrier()

    # every rank verifies the final values against the classic-math reference
    exp, _ = _expected_after_classic(init_rows, s, r, o, negs, cfg.lr, cfg.eps)
    keys = np.array(sorted(exp.keys()), dtype=np.int64)
    out = np.zeros((len(keys), 2 * DIM), dtype=np.float32)
    worker.pull(keys, out)
    for i, k in enumerate(keys):
        np.testing.assert_allclose(out[i], exp[int(k)], rtol=2e-4, atol=2e-5,
                                   err_msg=f"rank {rank} key {k}")
    worker.barrier()
    worker.finalize()
    server.shutdown()


@pytest.mark.parametrize("use_intent", [False, True])
def test_fused_general_world2(use_intent):
    run_dist(2, _fused_general_worker, use_intent, timeout=180)


def test_fused_general_world3():
    """world=3: manager != owner routing is common (key % 3), so misses,
    forwards and the classic remainder all exercise multi-hop paths."""
    run_dist(3, _fused_general_worker, True, timeout=180)


@pytest.mark.gpu
@pytest.mark.parametrize("use_intent", [False, True])
def test_fused_general_world2_gpu(use_intent):
    run_dist(2, _fused_general_worker, use_intent, "cuda:0", timeout=180)


def _w2v_mf_general_worker(rank, world):
    """w2v + MF models through the world>1 fused-general path (the same
    resolve/compact machinery as KGE, wired via train_*_fused
    force_general): losses stay finite and decrease, remote samples
    route through the classic path."""
    import adapm_amd
    from adapm_amd.models.mf import MF, MFConfig, make_synthetic_ratings
    from adapm_amd.models.word2vec import W2VConfig, Word2Vec, make_synthetic_sentences

    adapm_amd.setup(num_keys=2 * 64, num_threads=1, device="cpu",
                    max_sync_per_sec=4000.0)
    cfg = W2VConfig(vocab_size=64, dim=8, negative=2, window=3)
    server = adapm_amd.Server(cfg.row)
    worker = adapm_amd.Worker(0, server)
    model = Word2Vec(cfg, server, worker)
    model.init_embeddings()
    worker.barrier()
    sents = make_synthetic_sentences(40, 64, seed=rank)
    ctr, ctx = model.pairs_from_sentences(sents)
    losses = []
    for _ in range(6):
        losses.append(model.train_pairs_fused(ctr[:256], ctx[:256], sync_loss=True,
                                              force_general=True))
    model.drain()
    assert all(np.isfinite(l) for l in losses), losses
    # a step that deferred every sample reports 0.0 (nothing processed):
    # at world=2 the syn0/syn1 parity keying makes pairs never all-local
    # until replication kicks in, so skip empty steps
    nz = [l for l in losses if l > 0]
    # skip the first processed step: it covers only the biased local
    # subset (the deferred remainder lands in step 2)
    assert len(nz) >= 3 and nz[-1] < nz[1], losses
    worker.barrier()
    worker.finalize()
    server.shutdown()


def test_w2v_general_world2():
    run_dist(2, _w2v_mf_general_worker, timeout=180)


def _mf_general_worker(rank, world):
    import adapm_amd
    from adapm_amd.models.mf import MF, MFConfig, make_synthetic_ratings

    adapm_amd.setup(num_keys=80 + 40, num_threads=1, device="cpu",
                    max_sync_per_sec=4000.0)
    cfg = MFConfig(num_rows=80, num_cols=40, rank=8, lr=0.05)
    server = adapm_amd.Server(cfg.row)
    worker = adapm_amd.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()
    worker.barrier()
    rows, cols, ratings = make_synthetic_ratings(600, 80, 40, seed=3 + rank)
    losses = []
    for _ in range(8):
        losses.append(model.train_batch_fused(rows[:256], cols[:256], ratings[:256],
                                              sync_loss=True, force_general=True))
    model.drain()
    assert all(np.isfinite(l) for l in losses), losses
    nz = [l for l in losses if l > 0]
    assert len(nz) >= 3 and nz[-1] < nz[1], losses
    worker.barrier()
    worker.finalize()
    server.shutdown()


def test_mf_general_world2():
    run_dist(2, _mf_general_worker, timeout=180)
