"""Knowledge-graph embeddings (ComplEx) on the adaptive parameter manager.

Rebuild of the reference app (reference apps/knowledge_graph_embeddings.cc):
  - value layout: [embedding(dim) | AdaGrad accumulators(dim)] per key
    (reference entity_vector_length = 2*dim, :1243-1254)
  - entity keys [0, E), relation keys [E, E+R)
  - negatives via PrepareSample/PullSample (reference train() 437-531)
  - intent look-ahead: announce the keys of future batches
    `lookahead` batches ahead with per-batch clocks (reference
    signal_intent_ahead, :1059-1123)
  - score/grad/AdaGrad: one fused device kernel per batch
    (kernels_hip.hip k_kge_step; reference :832-858, 415-435)
  - eval: MRR / Hits@k by scoring candidate sets (reference :544-712)
  - checkpoint: WaitSync -> Barrier -> rank-0 full pull -> save
    (reference :327-401)
"""
from __future__ import annotations

import dataclasses
import os
import time
from collections import defaultdict
from typing import Optional

import numpy as np
import torch

_PHASE_TIMING = os.environ.get("ADAPM_PHASE_TIMING", "0") == "1"

import adapm_amd
from adapm_amd import _C


@dataclasses.dataclass
class ComplExConfig:
    num_entities: int = 1_000_000
    num_relations: int = 1_000
    dim: int = 512            # embedding dim (complex dim = dim/2)
    neg_samples: int = 16     # o-side negatives per positive
    batch_size: int = 4096
    lr: float = 0.1
    eps: float = 1e-6
    lookahead: int = 4        # batches of intent signaled ahead
    init_scale: float = 0.1
    seed: int = 42

    @property
    def num_keys(self) -> int:
        return self.num_entities + self.num_relations

    @property
    def row(self) -> int:
        return 2 * self.dim


class ComplEx:
    def __init__(self, cfg: ComplExConfig, server: "adapm_amd.Server",
                 worker: "adapm_amd.Worker"):
        self.cfg = cfg
        self.server = server
        self.worker = worker
        self.dev = server.rt.device
        self.rank = server.rt.rank
        self.world = server.rt.world
        self.rng = np.random.default_rng(cfg.seed + self.rank)
        self._pending = []
        self._deferred = None  # fused-general: missed samples retried next step
        self.phase_times = defaultdict(float)

    def _ph(self, name, t0):
        if _PHASE_TIMING:
            if self.dev.type == "cuda":
                torch.cuda.synchronize()
            t = time.perf_counter()
            self.phase_times[name] += t - t0
            return t
        return t0

    # ------------------------------------------------------------ init

    def init_embeddings(self):
        """Each rank initializes the keys it manages (key % world == rank)
        with set(); AdaGrad accumulators start at 0."""
        cfg = self.cfg
        chunk = max(1, 2 ** 25 // cfg.row)  # ~128MB of values per set
        my_keys = np.arange(self.rank, cfg.num_keys, self.world, dtype=np.int64)
        for i in range(0, len(my_keys), chunk):
            ks = my_keys[i:i + chunk]
            vals = torch.zeros(len(ks), cfg.row, dtype=torch.float32, device=self.dev)
            vals[:, :cfg.dim].normal_(0.0, cfg.init_scale)
            self.worker.set(ks, vals)
        self.worker.wait_sync()
        self.worker.barrier()

    # ------------------------------------------------------------ train

    def keys_of(self, triples: np.ndarray):
        s = triples[:, 0].astype(np.int64)
        r = (self.cfg.num_entities + triples[:, 1]).astype(np.int64)
        o = triples[:, 2].astype(np.int64)
        return s, r, o

    def signal_intent(self, triples: np.ndarray, start: int, end: int = 0):
        s, r, o = self.keys_of(triples)
        self.worker.intent(np.concatenate([s, r, o]), start, end)

    def _draw_negatives(self, B):
        cfg = self.cfg
        w = self.worker
        # negatives via the sampling manager (reference PrepareSample path)
        if self.server.sampling is not None:
            sid = w.prepare_sample(B * cfg.neg_samples, w.current_clock(),
                                   w.current_clock() + 2)
            neg_keys = self.server.sampling.pull(w, sid, B * cfg.neg_samples)
            w.finish_sample(sid)
        else:
            neg_keys = self.rng.integers(0, cfg.num_entities, size=B * cfg.neg_samples,
                                         dtype=np.int64)
        return neg_keys

    def train_batch(self, triples: np.ndarray, async_push: bool = True,
                    sync_loss: bool = True, neg_keys: np.ndarray = None):
        """One training step over B positive triples. Returns mean loss
        (float if sync_loss else a device tensor, no host sync)."""
        cfg = self.cfg
        w = self.worker
        B = len(triples)
        t0 = time.perf_counter() if _PHASE_TIMING else 0.0
        s_keys, r_keys, o_keys = self.keys_of(triples)
        if neg_keys is None:
            neg_keys = self._draw_negatives(B)
        t0 = self._ph("sample", t0)

        # one fused pull of all rows: [s | r | o | neg]
        dev = self.dev
        row = cfg.row
        NN = B * cfg.neg_samples
        all_keys = np.concatenate([s_keys, r_keys, o_keys, neg_keys])
        all_v = torch.empty((3 * B + NN) * row, dtype=torch.float32, device=dev)
        ts = w.pull(all_keys, all_v, async_=True)
        w.wait(ts)
        t0 = self._ph("pull", t0)

        s_v = all_v[: B * row].view(B, row)
        r_v = all_v[B * row: 2 * B * row].view(B, row)
        o_v = all_v[2 * B * row: 3 * B * row].view(B, row)
        n_v = all_v[3 * B * row:].view(NN, row)

        all_d = torch.empty_like(all_v)
        ds = all_d[: B * row].view(B, row)
        dr = all_d[B * row: 2 * B * row].view(B, row)
        do = all_d[2 * B * row: 3 * B * row].view(B, row)
        dn = all_d[3 * B * row:].view(NN, row)
        loss = torch.empty(B, dtype=torch.float32, device=dev)
        _C.kge_complex_step(s_v, r_v, o_v, n_v, ds, dr, do, dn, loss,
                            cfg.neg_samples, cfg.dim, cfg.lr, cfg.eps)
        t0 = self._ph("kernel", t0)

        pt = w.push(all_keys, all_d, async_=True)
        if pt != -1:
            self._pending.append(pt)
        if not async_push:
            w.wait(pt)
        # bounded async: cap outstanding pushes
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        t0 = self._ph("push", t0)
        if sync_loss:
            out = float(loss.mean().item())
            self._ph("loss_sync", t0)
            return out
        return loss

    def train_batch_fused(self, triples: np.ndarray, sync_loss: bool = False,
                          force_general: bool = False):
        """Fused slab-direct step: kernels read rows straight from the
        HBM slab and accumulate the AdaGrad deltas back — no pull/push
        buffers. world==1 with the identity layout uses the zero-host-
        work path (Server.kge_step_fused); otherwise the general path
        (kge_step_fused_general) resolves slab offsets in a host pass,
        runs the fused kernel on the samples whose keys are all local
        (>95% after intent-driven relocation) and routes the rest
        through the classic pull/kernel/push path. force_general also
        enables the general path on the CPU store (the gloo test tier)."""
        cfg = self.cfg
        w = self.worker
        raw = self.server.raw
        if self.dev.type != "cuda" and not force_general:
            return self.train_batch(triples, sync_loss=sync_loss)
        B = len(triples)
        s_keys, r_keys, o_keys = self.keys_of(triples)
        neg_keys = np.ascontiguousarray(self._draw_negatives(B), dtype=np.int64)

        if self.world == 1 and not force_general and raw.layout_identity():
            loss = raw.kge_step_fused(
                torch.from_numpy(s_keys), torch.from_numpy(r_keys), torch.from_numpy(o_keys),
                torch.from_numpy(neg_keys), cfg.neg_samples, cfg.dim, cfg.lr, cfg.eps)
            return float(loss.mean().item()) if sync_loss else loss

        # Missed (any-key-remote) samples would block THIS step on a
        # sync round-trip through the classic path. Intent was signaled
        # `lookahead` steps ago, so a miss is usually a straggler whose
        # replica lands within a round — DEFER it one step and retry
        # fused (second miss goes classic immediately, so nothing starves).
        negs2 = neg_keys.reshape(B, cfg.neg_samples)
        if self._deferred is not None:
            dtr, dng = self._deferred
            self._deferred = None
            n_def = len(dtr)
            triples = np.concatenate([dtr, triples])
            negs2 = np.concatenate([dng, negs2])
            s_keys, r_keys, o_keys = self.keys_of(triples)
            neg_keys = negs2.reshape(-1)
        else:
            n_def = 0

        loss, missed = raw.kge_step_fused_general(
            torch.from_numpy(s_keys), torch.from_numpy(r_keys), torch.from_numpy(o_keys),
            torch.from_numpy(np.ascontiguousarray(neg_keys)), cfg.neg_samples, cfg.dim,
            cfg.lr, cfg.eps)
        if missed.numel() == 0:
            return float(loss.mean().item()) if sync_loss else loss
        midx = missed.numpy()
        old = midx[midx < n_def]      # second miss: classic now
        fresh = midx[midx >= n_def]   # first miss: retry fused next step
        if len(fresh):
            self._deferred = (triples[fresh], negs2[fresh])
        if len(old):
            mloss = self.train_batch(triples[old], sync_loss=False,
                                     neg_keys=negs2[old].reshape(-1))
            if not torch.is_tensor(mloss):
                mloss = torch.tensor([mloss])
            loss = torch.cat([loss, mloss.to(loss.device)])
        if loss.numel() == 0:
            return 0.0 if sync_loss else loss
        return float(loss.mean().item()) if sync_loss else loss

    def drain(self):
        if self._deferred is not None:
            # flush straggler samples through the classic path so an
            # epoch boundary never drops work
            dtr, dng = self._deferred
            self._deferred = None
            self.train_batch(dtr, sync_loss=False, neg_keys=dng.reshape(-1))
        for t in self._pending:
            self.worker.wait(t)
        self._pending.clear()

    # ---------------------------------------------- prefetch pipeline

    def prefetch(self, triples: np.ndarray):
        """Start the pulls for a future batch (bounded async, like the
        reference's max_concurrent_loops pipelining): sample negatives,
        allocate buffers, issue one async pull. Returns a handle for
        train_prefetched."""
        cfg = self.cfg
        w = self.worker
        B = len(triples)
        s_keys, r_keys, o_keys = self.keys_of(triples)
        if self.server.sampling is not None:
            sid = w.prepare_sample(B * cfg.neg_samples, w.current_clock(),
                                   w.current_clock() + 2)
            neg_keys = self.server.sampling.pull(w, sid, B * cfg.neg_samples)
            w.finish_sample(sid)
        else:
            neg_keys = self.rng.integers(0, cfg.num_entities, size=B * cfg.neg_samples,
                                         dtype=np.int64)
        all_keys = np.concatenate([s_keys, r_keys, o_keys, neg_keys])
        all_v = torch.empty(len(all_keys) * cfg.row, dtype=torch.float32, device=self.dev)
        ts = w.pull(all_keys, all_v, async_=True)
        return (B, all_keys, all_v, ts)

    def train_prefetched(self, handle, sync_loss: bool = False):
        """Finish a prefetched batch: wait for the pull, run the fused
        kernel, push the deltas."""
        cfg = self.cfg
        w = self.worker
        B, all_keys, all_v, ts = handle
        w.wait(ts)
        row = cfg.row
        NN = B * cfg.neg_samples
        s_v = all_v[: B * row].view(B, row)
        r_v = all_v[B * row: 2 * B * row].view(B, row)
        o_v = all_v[2 * B * row: 3 * B * row].view(B, row)
        n_v = all_v[3 * B * row:].view(NN, row)
        all_d = torch.empty_like(all_v)
        loss = torch.empty(B, dtype=torch.float32, device=self.dev)
        _C.kge_complex_step(s_v, r_v, o_v, n_v, all_d[: B * row].view(B, row),
                            all_d[B * row: 2 * B * row].view(B, row),
                            all_d[2 * B * row: 3 * B * row].view(B, row),
                            all_d[3 * B * row:].view(NN, row), loss,
                            cfg.neg_samples, cfg.dim, cfg.lr, cfg.eps)
        pt = w.push(all_keys, all_d, async_=True)
        if pt != -1:
            self._pending.append(pt)
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        return float(loss.mean().item()) if sync_loss else loss

    # ------------------------------------------------------------ eval

    @torch.no_grad()
    def evaluate_full(self, triples: np.ndarray, hits_at=(1, 3, 10),
                      chunk: int = 65536, filter_triples: np.ndarray = None) -> dict:
        """Rank the true object among ALL entities, chunked (the reference
        evaluates against every entity, knowledge_graph_embeddings.cc:
        716-774 — there with OpenMP, here with the batched scoring
        kernel).

        filter_triples: known-true (s, r, o) triples (train+valid+test).
        When given, the FILTERED rank excludes other true objects of the
        same (s, r) from the competitors (reference computes filtered and
        raw ranks, knowledge_graph_embeddings.cc:544-712): `mrr`/`mr`/
        `hits@k` become the filtered metrics and `*_raw` carry the raw
        ones."""
        cfg = self.cfg
        w = self.worker
        B = len(triples)
        s_keys, r_keys, o_keys = self.keys_of(triples)
        opts = dict(dtype=torch.float32, device=self.dev)
        s_v = torch.empty(B, cfg.row, **opts)
        r_v = torch.empty(B, cfg.row, **opts)
        o_v = torch.empty(B, cfg.row, **opts)
        w.pull(s_keys, s_v)
        w.pull(r_keys, r_v)
        w.pull(o_keys, o_v)
        dc = cfg.dim // 2
        sr_re = s_v[:, :dc] * r_v[:, :dc] - s_v[:, dc:cfg.dim] * r_v[:, dc:cfg.dim]
        sr_im = s_v[:, dc:cfg.dim] * r_v[:, :dc] + s_v[:, :dc] * r_v[:, dc:cfg.dim]
        true_scores = (sr_re * o_v[:, :dc] + sr_im * o_v[:, dc:cfg.dim]).sum(1, keepdim=True)

        better = torch.zeros(B, dtype=torch.float32, device=self.dev)
        c_v = torch.empty(chunk, cfg.row, **opts)
        scores = torch.empty(B, chunk, **opts)
        for e0 in range(0, cfg.num_entities, chunk):
            n = min(chunk, cfg.num_entities - e0)
            cand = np.arange(e0, e0 + n, dtype=np.int64)
            cv = c_v[:n] if n == chunk else torch.empty(n, cfg.row, **opts)
            sv = scores[:, :n] if n == chunk else torch.empty(B, n, **opts)
            w.pull(cand, cv)
            _C.kge_complex_score(s_v, r_v, cv, sv, cfg.dim)
            better += (sv > true_scores).sum(1).float()
        rank_raw = 1 + better

        def metrics(rank, suffix=""):
            m = {f"mrr{suffix}": float((1.0 / rank).mean().item()),
                 f"mr{suffix}": float(rank.mean().item())}
            for h in hits_at:
                m[f"hits@{h}{suffix}"] = float((rank <= h).float().mean().item())
            return m

        if filter_triples is None:
            out = metrics(rank_raw)
        else:
            # filtered rank: other KNOWN-TRUE objects of the same (s, r)
            # do not count as competitors. Score only those few objects
            # and subtract the ones that out-ranked the true object.
            from collections import defaultdict

            true_objs = defaultdict(list)
            for fs, fr, fo in filter_triples:
                true_objs[(int(fs), int(fr))].append(int(fo))
            b_idx, obj_ids = [], []
            for b, (ts, tr, to) in enumerate(triples):
                for o2 in true_objs.get((int(ts), int(tr)), ()):
                    if o2 != int(to):
                        b_idx.append(b)
                        obj_ids.append(o2)
            filtered_better = torch.zeros_like(better)
            if b_idx:
                uniq, inv = np.unique(np.asarray(obj_ids, dtype=np.int64),
                                      return_inverse=True)
                u_v = torch.empty(len(uniq), cfg.row, **opts)
                w.pull(uniq, u_v)
                bi = torch.as_tensor(b_idx, dtype=torch.long, device=self.dev)
                ov2 = u_v[torch.as_tensor(inv, dtype=torch.long, device=self.dev)]
                sc = (sr_re[bi] * ov2[:, :dc] + sr_im[bi] * ov2[:, dc:cfg.dim]).sum(1)
                beats = (sc > true_scores[bi, 0]).float()
                filtered_better.scatter_add_(0, bi, beats)
            rank_filt = 1 + better - filtered_better
            out = metrics(rank_filt)
            out.update(metrics(rank_raw, "_raw"))
        out["n"] = float(B)
        if self.world > 1:
            vec = torch.tensor([out["n"]] + [out[k] * out["n"] for k in sorted(out) if k != "n"])
            vec = w.allreduce(vec)
            names = [k for k in sorted(out) if k != "n"]
            out = {k: float(vec[1 + i] / vec[0]) for i, k in enumerate(names)}
            out["n"] = float(vec[0])
        return out

    @torch.no_grad()
    def evaluate(self, triples: np.ndarray, num_candidates: int = 1000,
                 hits_at=(1, 3, 10)) -> dict:
        """Rank the true object among `num_candidates` random candidates
        (+ the true one). Aggregated over ranks via allreduce (reference
        eval: knowledge_graph_embeddings.cc:544-712)."""
        cfg = self.cfg
        w = self.worker
        B = len(triples)
        s_keys, r_keys, o_keys = self.keys_of(triples)
        cand = self.rng.integers(0, cfg.num_entities, size=num_candidates, dtype=np.int64)

        opts = dict(dtype=torch.float32, device=self.dev)
        s_v = torch.empty(B, cfg.row, **opts)
        r_v = torch.empty(B, cfg.row, **opts)
        o_v = torch.empty(B, cfg.row, **opts)
        c_v = torch.empty(num_candidates, cfg.row, **opts)
        w.pull(s_keys, s_v)
        w.pull(r_keys, r_v)
        w.pull(o_keys, o_v)
        w.pull(cand, c_v)

        cand_scores = torch.empty(B, num_candidates, **opts)
        _C.kge_complex_score(s_v, r_v, c_v, cand_scores, cfg.dim)
        true_scores = torch.empty(B, 1, **opts)
        # score the true o: reuse the kernel with E=1 per row via batched diag
        # (cheap path: einsum on device)
        dc = cfg.dim // 2
        sr_re = s_v[:, :dc] * r_v[:, :dc] - s_v[:, dc:cfg.dim] * r_v[:, dc:cfg.dim]
        sr_im = s_v[:, dc:cfg.dim] * r_v[:, :dc] + s_v[:, :dc] * r_v[:, dc:cfg.dim]
        true_scores[:, 0] = (sr_re * o_v[:, :dc] + sr_im * o_v[:, dc:cfg.dim]).sum(1)

        rank = 1 + (cand_scores > true_scores).sum(1).float()
        out = {
            "mrr": float((1.0 / rank).mean().item()),
            "mr": float(rank.mean().item()),
            "n": float(B),
        }
        for h in hits_at:
            out[f"hits@{h}"] = float((rank <= h).float().mean().item())
        # aggregate across ranks (weighted by n)
        if self.world > 1:
            vec = torch.tensor([out["n"]] + [out[k] * out["n"] for k in sorted(out) if k != "n"])
            vec = w.allreduce(vec)
            names = [k for k in sorted(out) if k != "n"]
            out = {k: float(vec[1 + i] / vec[0]) for i, k in enumerate(names)}
            out["n"] = float(vec[0])
        return out

    # ------------------------------------------------------------ checkpoint

    def save_checkpoint(self, path: str, chunk_keys: int = 65536):
        """rank 0 pulls the full model (incl. AdaGrad state) and writes an
        .npz (reference pull_full_model + write_checkpoint, :209-231,327-401)."""
        self.drain()
        self.worker.wait_sync()
        self.worker.barrier()
        if self.rank == 0:
            cfg = self.cfg
            out = np.empty((cfg.num_keys, cfg.row), dtype=np.float32)
            for i in range(0, cfg.num_keys, chunk_keys):
                ks = np.arange(i, min(i + chunk_keys, cfg.num_keys), dtype=np.int64)
                buf = np.zeros((len(ks), cfg.row), dtype=np.float32)
                self.worker.pull(ks, buf)
                out[i:i + len(ks)] = buf
            np.savez(path, values=out,
                     num_entities=cfg.num_entities, num_relations=cfg.num_relations,
                     dim=cfg.dim)
        self.worker.barrier()

    def load_checkpoint(self, path: str, chunk_keys: int = 65536):
        """rank 0 reads and set()s the full model."""
        if self.rank == 0:
            data = np.load(path)
            vals = data["values"]
            for i in range(0, len(vals), chunk_keys):
                ks = np.arange(i, min(i + chunk_keys, len(vals)), dtype=np.int64)
                self.worker.set(ks, np.ascontiguousarray(vals[i:i + len(ks)]))
        self.worker.wait_sync()
        self.worker.barrier()


def make_synthetic_triples(n: int, num_entities: int, num_relations: int,
                           seed: int = 0) -> np.ndarray:
    rng = np.random.default_rng(seed)
    return np.stack([
        rng.integers(0, num_entities, size=n),
        rng.integers(0, num_relations, size=n),
        rng.integers(0, num_entities, size=n),
    ], axis=1).astype(np.int64)


def main():
    """CLI (rebuild of the reference app binary,
    apps/knowledge_graph_embeddings.cc CLI): train ComplEx on a synthetic
    graph, report loss + eval, optionally checkpoint."""
    import argparse
    import time

    import adapm_amd as _a

    ap = argparse.ArgumentParser()
    ap.add_argument("--entities", type=int, default=100_000)
    ap.add_argument("--relations", type=int, default=100)
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--neg", type=int, default=8)
    ap.add_argument("--triples", type=int, default=1_000_000)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--lr", type=float, default=0.1)
    ap.add_argument("--eval-every", type=int, default=1)
    ap.add_argument("--eval-triples", type=int, default=1024)
    ap.add_argument("--eval-full", action="store_true",
                    help="rank against ALL entities (reference eval)")
    ap.add_argument("--checkpoint", type=str, default="")
    ap.add_argument("--device", type=str, default=None)
    a = ap.parse_args()

    _a.setup(num_keys=a.entities + a.relations, num_threads=1, device=a.device)
    server = _a.Server(2 * a.dim)
    server.enable_sampling_support("local", True, "uniform", 0, a.entities)
    worker = _a.Worker(0, server)
    cfg = ComplExConfig(num_entities=a.entities, num_relations=a.relations, dim=a.dim,
                        neg_samples=a.neg, batch_size=a.batch, lr=a.lr)
    model = ComplEx(cfg, server, worker)
    model.init_embeddings()
    rank = server.my_rank()
    world = server.rt.world
    triples = make_synthetic_triples(a.triples // world, a.entities, a.relations,
                                     seed=100 + rank)
    for ep in range(a.epochs):
        t0 = time.time()
        losses = []
        for i in range(0, len(triples), a.batch):
            b = triples[i:i + a.batch]
            model.signal_intent(b, worker.current_clock() + 1, worker.current_clock() + 3)
            losses.append(model.train_batch(b, sync_loss=True))
            worker.advance_clock()
        model.drain()
        total = worker.allreduce(float(np.mean(losses)))
        if rank == 0:
            print(f"[kge] epoch {ep}: loss {total / world:.4f} ({time.time()-t0:.1f}s)")
        if a.eval_every and (ep + 1) % a.eval_every == 0:
            ev = (model.evaluate_full(triples[:a.eval_triples], filter_triples=triples)
                  if a.eval_full else model.evaluate(triples[:a.eval_triples]))
            if rank == 0:
                print(f"[kge] eval: {ev}")
    if a.checkpoint:
        model.save_checkpoint(a.checkpoint)
        if rank == 0:
            print(f"[kge] checkpoint -> {a.checkpoint}")
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()


class Rescal:
    """RESCAL scorer (reference knowledge_graph_embeddings.cc:895-922):
    psi = e_s^T R e_o with a D x D matrix per relation. Exercises the
    store's NON-UNIFORM value lengths: entity rows 2*D floats, relation
    rows 2*D*D floats."""

    def __init__(self, cfg: ComplExConfig, server, worker):
        self.cfg = cfg
        self.server = server
        self.worker = worker
        self.dev = server.rt.device
        self.rank = server.rt.rank
        self.world = server.rt.world
        self.rng = np.random.default_rng(cfg.seed + self.rank)
        self._pending = []

    @staticmethod
    def value_lengths(num_entities, num_relations, dim):
        lens = np.full(num_entities + num_relations, 2 * dim, dtype=np.int64)
        lens[num_entities:] = 2 * dim * dim
        return lens

    def init_embeddings(self):
        cfg = self.cfg
        my_keys = np.arange(self.rank, cfg.num_keys, self.world, dtype=np.int64)
        ents = my_keys[my_keys < cfg.num_entities]
        rels = my_keys[my_keys >= cfg.num_entities]
        chunk = max(1, 2 ** 24 // (2 * cfg.dim))
        for i in range(0, len(ents), chunk):
            ks = ents[i:i + chunk]
            v = torch.zeros(len(ks), 2 * cfg.dim, dtype=torch.float32, device=self.dev)
            v[:, :cfg.dim].normal_(0, cfg.init_scale)
            self.worker.set(ks, v)
        rchunk = max(1, 2 ** 24 // (2 * cfg.dim * cfg.dim))
        for i in range(0, len(rels), rchunk):
            ks = rels[i:i + rchunk]
            v = torch.zeros(len(ks), 2 * cfg.dim * cfg.dim, dtype=torch.float32,
                            device=self.dev)
            v[:, :cfg.dim * cfg.dim].normal_(0, cfg.init_scale / cfg.dim ** 0.5)
            self.worker.set(ks, v)
        self.worker.wait_sync()
        self.worker.barrier()

    def train_batch(self, triples: np.ndarray, sync_loss: bool = True,
                    grouped: bool = None):
        """One RESCAL step. grouped=True (default on GPU): sort the batch
        by relation, pull each distinct relation matrix ONCE and run the
        three dim^2 products as MFMA-tiled grouped GEMMs
        (rescal_step_grouped) — the reference's per-triple scalar loops
        (knowledge_graph_embeddings.cc:895-922) are GEMM-shaped once
        triples share R. Also cuts the pull/push volume for R from
        B x 2D^2 to G x 2D^2. AdaGrad on dR uses the group-summed
        gradient (minibatch semantics); the classic path transforms each
        triple's outer product separately (both from the same pulled
        accumulator snapshot)."""
        cfg = self.cfg
        w = self.worker
        B = len(triples)
        D = cfg.dim
        if grouped is None:
            grouped = self.dev.type == "cuda" and D % 4 == 0
        if grouped:
            order = np.argsort(triples[:, 1], kind="stable")
            triples = triples[order]
        s_keys = triples[:, 0].astype(np.int64)
        r_keys = (cfg.num_entities + triples[:, 1]).astype(np.int64)
        o_keys = triples[:, 2].astype(np.int64)
        if self.server.sampling is not None:
            sid = w.prepare_sample(B * cfg.neg_samples, w.current_clock(), w.current_clock() + 2)
            neg_keys = self.server.sampling.pull(w, sid, B * cfg.neg_samples)
            w.finish_sample(sid)
        else:
            neg_keys = self.rng.integers(0, cfg.num_entities, size=B * cfg.neg_samples,
                                         dtype=np.int64)
        opts = dict(dtype=torch.float32, device=self.dev)
        s_v = torch.empty(B, 2 * D, **opts)
        o_v = torch.empty(B, 2 * D, **opts)
        n_v = torch.empty(B * cfg.neg_samples, 2 * D, **opts)

        if grouped:
            uniq_r, starts_np = np.unique(r_keys, return_index=True)
            starts = np.append(starts_np, B).astype(np.int32)
            r_v = torch.empty(len(uniq_r), 2 * D * D, **opts)
            for kt, vt in ((s_keys, s_v), (uniq_r, r_v), (o_keys, o_v), (neg_keys, n_v)):
                w.wait(w.pull(kt, vt, async_=True))
            ds, do, dn = (torch.empty_like(t) for t in (s_v, o_v, n_v))
            drl = torch.empty_like(r_v)
            loss = _C.rescal_step_grouped(s_v, r_v, o_v, n_v, ds, drl, do, dn,
                                          torch.from_numpy(starts), cfg.neg_samples, D,
                                          cfg.lr, cfg.eps)
            push_sets = ((s_keys, ds), (uniq_r, drl), (o_keys, do), (neg_keys, dn))
        else:
            r_v = torch.empty(B, 2 * D * D, **opts)
            for kt, vt in ((s_keys, s_v), (r_keys, r_v), (o_keys, o_v), (neg_keys, n_v)):
                w.wait(w.pull(kt, vt, async_=True))
            ds, drl, do, dn = (torch.empty_like(t) for t in (s_v, r_v, o_v, n_v))
            loss = torch.empty(B, dtype=torch.float32, device=self.dev)
            _C.rescal_step(s_v, r_v, o_v, n_v, ds, drl, do, dn, loss,
                           cfg.neg_samples, D, cfg.lr, cfg.eps)
            push_sets = ((s_keys, ds), (r_keys, drl), (o_keys, do), (neg_keys, dn))
        for kt, dt in push_sets:
            pt = w.push(kt, dt, async_=True)
            if pt != -1:
                self._pending.append(pt)
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        return float(loss.mean().item()) if sync_loss else loss

    def drain(self):
        for t in self._pending:
            self.worker.wait(t)
        self._pending.clear()
