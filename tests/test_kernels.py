"""App-kernel numerics vs plain PyTorch fp32 references (autograd).

CPU tests run everywhere; the @gpu twins run the HIP kernels on an MI355X
and compare against the same torch references.
"""
import numpy as np
import pytest
import torch

from adapm_amd import _C


# ------------------------------------------------------------ references

def torch_kge_ref(s, r, o, neg, N, D, lr, eps):
    """ComplEx logistic loss with o-side negatives; returns expected push
    deltas [delta_emb | grad^2] for s, r, o, neg rows."""
    B = s.shape[0]
    dc = D // 2

    def leaf(x):
        t = x[:, :D].clone().detach().requires_grad_(True)
        return t

    se, re_, oe, ne = leaf(s), leaf(r), leaf(o), leaf(neg)

    def score(sv, rv, ov):
        sre, sim = sv[:, :dc], sv[:, dc:]
        rre, rim = rv[:, :dc], rv[:, dc:]
        ore, oim = ov[:, :dc], ov[:, dc:]
        ure = sre * rre - sim * rim
        uim = sim * rre + sre * rim
        return (ure * ore + uim * oim).sum(1)

    pos = score(se, re_, oe)
    loss = torch.nn.functional.softplus(-pos).sum()
    nv = ne.view(B, N, D)
    for j in range(N):
        neg_score = score(se, re_, nv[:, j, :])
        loss = loss + torch.nn.functional.softplus(neg_score).sum()
    loss.backward()

    def delta(x_full, leaf_t):
        g = leaf_t.grad
        G = x_full[:, D:] + g * g
        d = torch.cat([-lr * g / torch.sqrt(G + eps), g * g], dim=1)
        return d

    return delta(s, se), delta(r, re_), delta(o, oe), delta(neg, ne)


def run_kge(device, B=4, N=3, D=16, tol=2e-4):
    g = torch.Generator().manual_seed(0)
    mk = lambda n: torch.cat([torch.randn(n, D, generator=g) * 0.3,
                              torch.rand(n, D, generator=g) * 0.1], dim=1)
    s, r, o, neg = mk(B), mk(B), mk(B), mk(B * N)
    lr, eps = 0.05, 1e-6

    sd, rd, od, nd = [t.to(device) for t in (s, r, o, neg)]
    ds, dr, do, dn = [torch.empty_like(t) for t in (sd, rd, od, nd)]
    loss = torch.empty(B, dtype=torch.float32, device=device)
    _C.kge_complex_step(sd, rd, od, nd, ds, dr, do, dn, loss, N, D, lr, eps)

    eds, edr, edo, edn = torch_kge_ref(s, r, o, neg, N, D, lr, eps)
    for got, exp, name in [(ds, eds, "ds"), (dr, edr, "dr"), (do, edo, "do"), (dn, edn, "dn")]:
        got = got.cpu()
        err = (got - exp).abs().max().item()
        assert err < tol, f"{name} max err {err}"
    assert torch.isfinite(loss).all()


def test_kge_step_cpu():
    run_kge("cpu")


def test_kge_step_cpu_large_dim():
    run_kge("cpu", B=2, N=2, D=512)


@pytest.mark.gpu
def test_kge_step_gpu():
    run_kge("cuda:0")


@pytest.mark.gpu
def test_kge_step_gpu_dim512():
    run_kge("cuda:0", B=8, N=8, D=512)


@pytest.mark.gpu
def test_kge_step_gpu_dim1024():
    run_kge("cuda:0", B=4, N=4, D=1024)


# ------------------------------------------------------------ scoring

def run_kge_score(device, B=3, E=5, D=32):
    g = torch.Generator().manual_seed(1)
    mk = lambda n: torch.cat([torch.randn(n, D, generator=g) * 0.3,
                              torch.zeros(n, D)], dim=1)
    s, r, cand = mk(B), mk(B), mk(E)
    dc = D // 2
    sre, sim = s[:, :dc], s[:, dc:D]
    rre, rim = r[:, :dc], r[:, dc:D]
    ure = sre * rre - sim * rim
    uim = sim * rre + sre * rim
    exp = ure @ cand[:, :dc].T + uim @ cand[:, dc:D].T

    sd, rd, cd = s.to(device), r.to(device), cand.to(device)
    scores = torch.empty(B, E, dtype=torch.float32, device=device)
    _C.kge_complex_score(sd, rd, cd, scores, D)
    assert (scores.cpu() - exp).abs().max().item() < 1e-4


def test_kge_score_cpu():
    run_kge_score("cpu")


@pytest.mark.gpu
def test_kge_score_gpu():
    run_kge_score("cuda:0", B=8, E=64, D=512)


# ------------------------------------------------------------ word2vec

def torch_w2v_ref(ctr, ctx, neg, N, D, lr, eps):
    B = ctr.shape[0]
    ce = ctr[:, :D].clone().detach().requires_grad_(True)
    xe = ctx[:, :D].clone().detach().requires_grad_(True)
    ne = neg[:, :D].clone().detach().requires_grad_(True)
    loss = torch.nn.functional.softplus(-(ce * xe).sum(1)).sum()
    nv = ne.view(B, N, D)
    for j in range(N):
        loss = loss + torch.nn.functional.softplus((ce * nv[:, j, :]).sum(1)).sum()
    loss.backward()

    def delta(full, leaf):
        g = leaf.grad
        G = full[:, D:] + g * g
        return torch.cat([-lr * g / torch.sqrt(G + eps), g * g], dim=1)

    return delta(ctr, ce), delta(ctx, xe), delta(neg, ne)


def run_w2v(device, B=4, N=3, D=24, tol=2e-4):
    g = torch.Generator().manual_seed(2)
    mk = lambda n: torch.cat([torch.randn(n, D, generator=g) * 0.3,
                              torch.rand(n, D, generator=g) * 0.1], dim=1)
    ctr, ctx, neg = mk(B), mk(B), mk(B * N)
    lr, eps = 0.025, 1e-6
    cd, xd, nd = [t.to(device) for t in (ctr, ctx, neg)]
    dc, dx, dn = [torch.empty_like(t) for t in (cd, xd, nd)]
    loss = torch.empty(B, dtype=torch.float32, device=device)
    _C.w2v_sgns_step(cd, xd, nd, dc, dx, dn, loss, N, D, lr, eps)
    edc, edx, edn = torch_w2v_ref(ctr, ctx, neg, N, D, lr, eps)
    for got, exp, name in [(dc, edc, "dctr"), (dx, edx, "dctx"), (dn, edn, "dneg")]:
        err = (got.cpu() - exp).abs().max().item()
        assert err < tol, f"{name} max err {err}"


def test_w2v_step_cpu():
    run_w2v("cpu")


@pytest.mark.gpu
def test_w2v_step_gpu():
    run_w2v("cuda:0", B=8, N=5, D=300)


# ------------------------------------------------------------ MF

def torch_mf_ref(w, h, x, R, lr, lam, eps):
    we = w[:, :R].clone().detach().requires_grad_(True)
    he = h[:, :R].clone().detach().requires_grad_(True)
    e = x - (we * he).sum(1)
    loss = (e * e).sum() + lam * ((we * we).sum() + (he * he).sum())
    loss.backward()

    def delta(full, leaf):
        g = leaf.grad
        G = full[:, R:] + g * g
        return torch.cat([-lr * g / torch.sqrt(G + eps), g * g], dim=1)

    return delta(w, we), delta(h, he)


def run_mf(device, B=6, R=32, tol=2e-4):
    g = torch.Generator().manual_seed(3)
    mk = lambda n: torch.cat([torch.randn(n, R, generator=g) * 0.3,
                              torch.rand(n, R, generator=g) * 0.1], dim=1)
    w, h = mk(B), mk(B)
    x = torch.randn(B, generator=g)
    lr, lam, eps = 0.01, 0.05, 1e-6
    wd, hd, xd = w.to(device), h.to(device), x.to(device)
    dw, dh = torch.empty_like(wd), torch.empty_like(hd)
    loss = torch.empty(B, dtype=torch.float32, device=device)
    _C.mf_update_step(wd, hd, xd, dw, dh, loss, R, lr, lam, eps)
    edw, edh = torch_mf_ref(w, h, x, R, lr, lam, eps)
    assert (dw.cpu() - edw).abs().max().item() < tol
    assert (dh.cpu() - edh).abs().max().item() < tol
    exp_loss = (x - (w[:, :R] * h[:, :R]).sum(1)) ** 2
    assert (loss.cpu() - exp_loss).abs().max().item() < 1e-4


def test_mf_step_cpu():
    run_mf("cpu")


@pytest.mark.gpu
def test_mf_step_gpu():
    run_mf("cuda:0", B=16, R=128)


# ------------------------------------------------------------ RESCAL

def torch_rescal_ref(s, r, o, neg, N, D, lr, eps):
    B = s.shape[0]
    se = s[:, :D].clone().detach().requires_grad_(True)
    Re = r[:, :D * D].clone().detach().requires_grad_(True)
    oe = o[:, :D].clone().detach().requires_grad_(True)
    ne = neg[:, :D].clone().detach().requires_grad_(True)
    R = Re.view(B, D, D)
    pos = torch.einsum("bi,bij,bj->b", se, R, oe)
    loss = torch.nn.functional.softplus(-pos).sum()
    nv = ne.view(B, N, D)
    for j in range(N):
        sc = torch.einsum("bi,bij,bj->b", se, R, nv[:, j, :])
        loss = loss + torch.nn.functional.softplus(sc).sum()
    loss.backward()

    def delta(full, leaf, L):
        g = leaf.grad.reshape(full.shape[0], L)
        G = full[:, L:] + g * g
        return torch.cat([-lr * g / torch.sqrt(G + eps), g * g], dim=1)

    return (delta(s, se, D), delta(r, Re, D * D), delta(o, oe, D), delta(neg, ne, D))


def run_rescal(device, B=3, N=2, D=16, tol=3e-4):
    g = torch.Generator().manual_seed(5)
    mke = lambda n: torch.cat([torch.randn(n, D, generator=g) * 0.3,
                               torch.rand(n, D, generator=g) * 0.1], dim=1)
    s, o, neg = mke(B), mke(B), mke(B * N)
    r = torch.cat([torch.randn(B, D * D, generator=g) * 0.1,
                   torch.rand(B, D * D, generator=g) * 0.1], dim=1)
    lr, eps = 0.05, 1e-6
    sd, rd, od, nd = (t.to(device) for t in (s, r, o, neg))
    ds, dr, do, dn = (torch.empty_like(t) for t in (sd, rd, od, nd))
    loss = torch.empty(B, dtype=torch.float32, device=device)
    from adapm_amd import _C as C2

    C2.rescal_step(sd, rd, od, nd, ds, dr, do, dn, loss, N, D, lr, eps)
    eds, edr, edo, edn = torch_rescal_ref(s, r, o, neg, N, D, lr, eps)
    for got, exp, name in [(ds, eds, "ds"), (dr, edr, "dR"), (do, edo, "do"), (dn, edn, "dn")]:
        err = (got.cpu() - exp).abs().max().item()
        assert err < tol, f"{name} max err {err}"


def test_rescal_cpu():
    run_rescal("cpu")


@pytest.mark.gpu
def test_rescal_gpu():
    run_rescal("cuda:0", B=8, N=4, D=128)


def test_rescal_model_cpu():
    """End-to-end RESCAL on the store with NON-UNIFORM value lengths."""
    import adapm_amd
    from adapm_amd.models.kge import ComplExConfig, Rescal, make_synthetic_triples

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    E, R, D = 200, 10, 16
    adapm_amd.setup(num_keys=E + R, num_threads=1, device="cpu")
    lens = Rescal.value_lengths(E, R, D)
    server = adapm_amd.Server(torch.from_numpy(lens))
    worker = adapm_amd.Worker(0, server)
    cfg = ComplExConfig(num_entities=E, num_relations=R, dim=D, neg_samples=2,
                        batch_size=64, lr=0.1)
    model = Rescal(cfg, server, worker)
    model.init_embeddings()
    triples = make_synthetic_triples(256, E, R, seed=2)
    losses = []
    for ep in range(6):
        for i in range(0, len(triples), 64):
            losses.append(model.train_batch(triples[i:i + 64]))
    model.drain()
    assert losses[-1] < losses[0], losses[:3] + losses[-3:]
    worker.finalize()
    server.shutdown()


@pytest.mark.gpu
def test_kge_fused_matches_classic():
    """Fused slab-direct step == classic pull/kernel/push on UNIQUE keys
    (duplicates are hogwild-updated by design)."""
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    E, R, D, N, B = 500, 64, 64, 3, 32
    adapm_amd.setup(num_keys=E + R, num_threads=1, device="cuda:0")
    server = adapm_amd.Server(2 * D)
    worker = adapm_amd.Worker(0, server)
    g = torch.Generator().manual_seed(0)
    init = torch.cat([torch.randn(E + R, D, generator=g) * 0.2,
                      torch.rand(E + R, D, generator=g) * 0.1], dim=1)
    worker.set(np.arange(E + R, dtype=np.int64), init.cuda())

    rng = np.random.default_rng(1)
    # all keys unique so fused == classic bit-for-bit (duplicates are
    # hogwild in the fused kernel by design)
    s_k = rng.choice(E, B, replace=False).astype(np.int64)
    r_k = (E + rng.choice(R, B, replace=False)).astype(np.int64)
    o_k = rng.choice(np.setdiff1d(np.arange(E), s_k), B, replace=False).astype(np.int64)
    neg_pool = np.setdiff1d(np.arange(E), np.concatenate([s_k, o_k]))
    n_k = rng.choice(neg_pool, B * N, replace=False).astype(np.int64)

    # classic path on a snapshot
    all_keys = np.concatenate([s_k, r_k, o_k, n_k])
    rows = torch.empty(len(all_keys), 2 * D, device="cuda")
    worker.pull(all_keys, rows)
    sv, rv, ov, nv = (rows[:B], rows[B:2 * B], rows[2 * B:3 * B], rows[3 * B:])
    from adapm_amd import _C as C3

    ds, dr, do, dn = (torch.empty_like(t) for t in (sv, rv, ov, nv))
    loss_c = torch.empty(B, device="cuda")
    C3.kge_complex_step(sv.contiguous(), rv.contiguous(), ov.contiguous(), nv.contiguous(),
                        ds, dr, do, dn, loss_c, N, D, 0.05, 1e-6)

    # fused path mutates the store
    loss_f = server.raw.kge_step_fused(torch.from_numpy(s_k), torch.from_numpy(r_k),
                                       torch.from_numpy(o_k), torch.from_numpy(n_k),
                                       N, D, 0.05, 1e-6)
    torch.cuda.synchronize()
    assert torch.allclose(loss_f, loss_c, atol=1e-4)

    after = torch.empty(len(all_keys), 2 * D, device="cuda")
    worker.pull(all_keys, after)
    torch.cuda.synchronize()
    expected = rows + torch.cat([ds, dr, do, dn])
    err = (after - expected).abs().max().item()
    assert err < 1e-4, f"fused store update mismatch: {err}"
    server.shutdown()


@pytest.mark.gpu
def test_w2v_fused_matches_classic():
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    V, D, N, B = 600, 48, 3, 24
    adapm_amd.setup(num_keys=2 * V, num_threads=1, device="cuda:0")
    server = adapm_amd.Server(2 * D)
    worker = adapm_amd.Worker(0, server)
    g = torch.Generator().manual_seed(3)
    init = torch.cat([torch.randn(2 * V, D, generator=g) * 0.2,
                      torch.rand(2 * V, D, generator=g) * 0.1], dim=1)
    worker.set(np.arange(2 * V, dtype=np.int64), init.cuda())

    rng = np.random.default_rng(4)
    # all keys unique (ctr on even keys, ctx+neg on odd keys)
    ctr_k = 2 * rng.choice(V, B, replace=False).astype(np.int64)
    rest = rng.choice(V, B + B * N, replace=False).astype(np.int64)
    ctx_k = 2 * rest[:B] + 1
    neg_k = 2 * rest[B:] + 1

    all_keys = np.concatenate([ctr_k, ctx_k, neg_k])
    rows = torch.empty(len(all_keys), 2 * D, device="cuda")
    worker.pull(all_keys, rows)
    cv, xv, nv = rows[:B], rows[B:2 * B], rows[2 * B:]
    from adapm_amd import _C as C

    dc, dx, dn = (torch.empty_like(t) for t in (cv, xv, nv))
    loss_c = torch.empty(B, device="cuda")
    C.w2v_sgns_step(cv.contiguous(), xv.contiguous(), nv.contiguous(),
                    dc, dx, dn, loss_c, N, D, 0.05, 1e-6)

    loss_f = server.raw.w2v_step_fused(torch.from_numpy(ctr_k), torch.from_numpy(ctx_k),
                                       torch.from_numpy(neg_k), N, D, 0.05, 1e-6)
    torch.cuda.synchronize()
    assert torch.allclose(loss_f, loss_c, atol=1e-4)
    after = torch.empty(len(all_keys), 2 * D, device="cuda")
    worker.pull(all_keys, after)
    torch.cuda.synchronize()
    expected = rows + torch.cat([dc, dx, dn])
    err = (after - expected).abs().max().item()
    assert err < 1e-4, f"fused w2v store update mismatch: {err}"
    server.shutdown()


@pytest.mark.gpu
def test_mf_fused_matches_classic():
    import adapm_amd

    adapm_amd._SETUP.clear()
    adapm_amd.runtime._RUNTIME = None
    NR, NC, R, B = 400, 300, 32, 48
    adapm_amd.setup(num_keys=NR + NC, num_threads=1, device="cuda:0")
    server = adapm_amd.Server(2 * R)
    worker = adapm_amd.Worker(0, server)
    g = torch.Generator().manual_seed(5)
    init = torch.cat([torch.rand(NR + NC, R, generator=g) * 0.1,
                      torch.full((NR + NC, R), 1.0)], dim=1)
    worker.set(np.arange(NR + NC, dtype=np.int64), init.cuda())

    rng = np.random.default_rng(6)
    w_k = rng.choice(NR, B, replace=False).astype(np.int64)
    h_k = (NR + rng.choice(NC, B, replace=False)).astype(np.int64)
    x = rng.normal(size=B).astype(np.float32)

    all_keys = np.concatenate([w_k, h_k])
    rows = torch.empty(2 * B, 2 * R, device="cuda")
    worker.pull(all_keys, rows)
    wv, hv = rows[:B], rows[B:]
    from adapm_amd import _C as C

    dw, dh = torch.empty_like(wv), torch.empty_like(hv)
    loss_c = torch.empty(B, device="cuda")
    C.mf_update_step(wv.contiguous(), hv.contiguous(),
                     torch.from_numpy(x).cuda(), dw, dh, loss_c, R, 0.05, 0.01, 1e-6)

    loss_f = server.raw.mf_step_fused(torch.from_numpy(w_k), torch.from_numpy(h_k),
                                      torch.from_numpy(x), R, 0.05, 0.01, 1e-6)
    torch.cuda.synchronize()
    assert torch.allclose(loss_f, loss_c, atol=1e-4)
    after = torch.empty(2 * B, 2 * R, device="cuda")
    worker.pull(all_keys, after)
    torch.cuda.synchronize()
    expected = rows + torch.cat([dw, dh])
    err = (after - expected).abs().max().item()
    assert err < 1e-4, f"fused mf store update mismatch: {err}"
    server.shutdown()


def test_mf_loss_kernel_cpu():
    """Dedicated MF loss reduction (reference apps/mf/loss.h): NZSL + L2
    vs a torch fp32 reference."""
    import adapm_amd
    from adapm_amd import _C

    g = torch.Generator().manual_seed(5)
    B, R, lam = 257, 16, 0.05
    w = torch.randn(B, 2 * R, generator=g)
    h = torch.randn(B, 2 * R, generator=g)
    x = torch.randn(B, generator=g)
    out2 = _C.mf_loss(w, h, x, R, lam)
    pred = (w[:, :R] * h[:, :R]).sum(1)
    se_ref = ((x - pred) ** 2).sum()
    reg_ref = lam * ((w[:, :R] ** 2).sum() + (h[:, :R] ** 2).sum())
    torch.testing.assert_close(out2[0], se_ref, rtol=1e-4, atol=1e-3)
    torch.testing.assert_close(out2[1], reg_ref, rtol=1e-4, atol=1e-3)


@pytest.mark.gpu
def test_mf_loss_kernel_gpu():
    import adapm_amd
    from adapm_amd import _C

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    g = torch.Generator().manual_seed(6)
    B, R, lam = 4097, 128, 0.02
    w = torch.randn(B, 2 * R, generator=g).cuda()
    h = torch.randn(B, 2 * R, generator=g).cuda()
    x = torch.randn(B, generator=g).cuda()
    out2 = _C.mf_loss(w, h, x, R, lam)
    pred = (w[:, :R] * h[:, :R]).sum(1)
    se_ref = ((x - pred) ** 2).sum()
    reg_ref = lam * ((w[:, :R] ** 2).sum() + (h[:, :R] ** 2).sum())
    torch.cuda.synchronize()
    torch.testing.assert_close(out2[0], se_ref, rtol=1e-3, atol=1e-1)
    torch.testing.assert_close(out2[1], reg_ref, rtol=1e-3, atol=1e-1)
