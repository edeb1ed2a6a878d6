"""MFMA (matrix-core) kernel numerics: the eval-scoring GEMM and the
grouped RESCAL step against plain-torch fp32 references. GPU-only (the
CPU tier uses the scalar reference kernels)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")


def _softplus(x):
    return torch.where(x > 20, x, torch.log1p(torch.exp(x)))


@requires_gpu
@pytest.mark.parametrize("B,E,D", [(37, 1000, 64), (16, 64, 512), (5, 129, 8)])
def test_mfma_scores_matches_torch(B, E, D):
    from adapm_amd import _C

    g = torch.Generator(device="cpu").manual_seed(42)
    s = torch.randn(B, 2 * D, generator=g).cuda()
    r = torch.randn(B, 2 * D, generator=g).cuda()
    cand = torch.randn(E, 2 * D, generator=g).cuda()
    scores = torch.empty(B, E, device="cuda")
    _C.kge_complex_score(s, r, cand, scores, D)
    dc = D // 2
    q_re = s[:, :dc] * r[:, :dc] - s[:, dc:D] * r[:, dc:D]
    q_im = s[:, dc:D] * r[:, :dc] + s[:, :dc] * r[:, dc:D]
    q = torch.cat([q_re, q_im], dim=1)
    ref = q @ cand[:, :D].T
    torch.cuda.synchronize()
    torch.testing.assert_close(scores, ref, rtol=2e-4, atol=2e-4)


@requires_gpu
def test_rescal_grouped_matches_reference():
    from adapm_amd import _C

    B, G, N, D = 21, 4, 3, 32
    lr, eps = 0.1, 1e-8
    g = torch.Generator(device="cpu").manual_seed(7)
    # triples sorted by relation; uneven group sizes
    counts = [2, 9, 4, 6]
    starts = np.cumsum([0] + counts).astype(np.int32)
    s = torch.randn(B, 2 * D, generator=g).abs_().add_(0.01)
    s[:, :D] = torch.randn(B, D, generator=g)
    rm = torch.randn(G, 2 * D * D, generator=g).abs_().add_(0.01)
    rm[:, :D * D] = torch.randn(G, D * D, generator=g) * 0.1
    o = torch.randn(B, 2 * D, generator=g).abs_().add_(0.01)
    o[:, :D] = torch.randn(B, D, generator=g)
    neg = torch.randn(B * N, 2 * D, generator=g).abs_().add_(0.01)
    neg[:, :D] = torch.randn(B * N, D, generator=g)

    sd, rmd, od, nd = (t.cuda() for t in (s, rm, o, neg))
    ds = torch.empty_like(sd)
    drl = torch.empty_like(rmd)
    do = torch.empty_like(od)
    dn = torch.empty_like(nd)
    loss = _C.rescal_step_grouped(sd, rmd, od, nd, ds, drl, do, dn,
                                  torch.from_numpy(starts), N, D, lr, eps)
    torch.cuda.synchronize()

    # ---- torch fp32 reference (same factorization + group-summed dR)
    ref_ds = torch.empty_like(s)
    ref_do = torch.empty_like(o)
    ref_dn = torch.empty_like(neg)
    ref_drl = torch.empty_like(rm)
    ref_loss = torch.empty(B)
    for gi in range(G):
        lo, hi = int(starts[gi]), int(starts[gi + 1])
        R = rm[gi, :D * D].view(D, D)
        dR_sum = torch.zeros(D, D)
        for b in range(lo, hi):
            es = s[b, :D]
            u = R.T @ es
            w = torch.zeros(D)
            for j in range(N + 1):
                ob = o[b] if j == 0 else neg[b * N + j - 1]
                dob = ref_do[b] if j == 0 else ref_dn[b * N + j - 1]
                y = 1.0 if j == 0 else -1.0
                dot = float(u @ ob[:D])
                c = -y * torch.sigmoid(torch.tensor(-y * dot)).item()
                if j == 0:
                    ref_loss[b] = _softplus(torch.tensor(-y * dot))
                else:
                    ref_loss[b] += _softplus(torch.tensor(-y * dot))
                w += c * ob[:D]
                gvec = c * u
                dob[:D] = -lr * gvec / torch.sqrt(ob[D:] + gvec * gvec + eps)
                dob[D:] = gvec * gvec
            gs = R @ w
            ref_ds[b, :D] = -lr * gs / torch.sqrt(s[b, D:] + gs * gs + eps)
            ref_ds[b, D:] = gs * gs
            dR_sum += torch.outer(es, w)
        gr = dR_sum.reshape(-1)
        ref_drl[gi, :D * D] = -lr * gr / torch.sqrt(rm[gi, D * D:] + gr * gr + eps)
        ref_drl[gi, D * D:] = gr * gr

    torch.testing.assert_close(loss.cpu(), ref_loss, rtol=2e-3, atol=2e-4)
    torch.testing.assert_close(do.cpu(), ref_do, rtol=2e-3, atol=2e-4)
    torch.testing.assert_close(dn.cpu(), ref_dn, rtol=2e-3, atol=2e-4)
    torch.testing.assert_close(ds.cpu(), ref_ds, rtol=2e-3, atol=2e-4)
    torch.testing.assert_close(drl.cpu(), ref_drl, rtol=2e-3, atol=2e-4)


@requires_gpu
def test_rescal_model_grouped_learns():
    """End-to-end RESCAL model on the store with the grouped path."""
    import adapm_amd
    from adapm_amd.models.kge import ComplExConfig, Rescal, make_synthetic_triples

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    E, R, D = 300, 6, 32
    adapm_amd.setup(num_keys=E + R, num_threads=1, device="cuda:0")
    lens = Rescal.value_lengths(E, R, D)
    server = adapm_amd.Server(torch.from_numpy(lens))
    worker = adapm_amd.Worker(0, server)
    cfg = ComplExConfig(num_entities=E, num_relations=R, dim=D, neg_samples=4,
                        batch_size=64, lr=0.1)
    model = Rescal(cfg, server, worker)
    model.init_embeddings()
    triples = make_synthetic_triples(256, E, R, seed=2)
    losses = []
    for _ in range(12):
        for i in range(0, len(triples), cfg.batch_size):
            losses.append(model.train_batch(triples[i:i + cfg.batch_size], grouped=True))
    model.drain()
    torch.cuda.synchronize()
    assert losses[-1] < losses[0] * 0.8, f"no learning: {losses[0]} -> {losses[-1]}"
    worker.finalize()
    server.shutdown()
