"""word2vec SGNS on the adaptive parameter manager.

Rebuild of the reference app (reference apps/word2vec.cc):
  - keys: syn0(word w) = 2w, syn1(w) = 2w+1 (reference :83-105)
  - value layout [embedding(dim) | AdaGrad accum(dim)]
  - negatives from the unigram^0.75 distribution (reference :125-146)
    via the sampling manager
  - sentence look-ahead intent prefetch (reference :563-618)
  - fused SGNS step kernel (kernels_hip.hip k_w2v_step; reference
    :679-745)
  - checkpoint: word2vec text-format embedding export (reference
    :367-416)
"""
from __future__ import annotations

import dataclasses

import numpy as np
import torch

import adapm_amd
from adapm_amd import _C


@dataclasses.dataclass
class W2VConfig:
    vocab_size: int = 1_000_000
    dim: int = 300
    window: int = 5
    negative: int = 5
    batch_pairs: int = 16384
    lr: float = 0.025
    eps: float = 1e-6
    lookahead_sentences: int = 100
    accum_init: float = 1.0  # AdaGrad accumulator start (0 diverges: first
                             # step would be lr*sign(g) per element)
    seed: int = 13

    @property
    def num_keys(self):
        return 2 * self.vocab_size

    @property
    def row(self):
        return 2 * self.dim


def syn0(words):
    return 2 * np.asarray(words, dtype=np.int64)


def syn1(words):
    return 2 * np.asarray(words, dtype=np.int64) + 1


class Word2Vec:
    def __init__(self, cfg: W2VConfig, server, worker):
        self.cfg = cfg
        self.server = server
        self.worker = worker
        self.dev = server.rt.device
        self.rank = server.rt.rank
        self.world = server.rt.world
        self.rng = np.random.default_rng(cfg.seed + self.rank)
        self._pending = []
        self._deferred = None  # fused-general: missed pairs retried next step
        self._keep_prob = None
        self.words = None  # real-corpus vocab strings (data_io.build_vocab)

    def init_embeddings(self):
        cfg = self.cfg
        chunk = max(1, 2 ** 25 // cfg.row)
        my_keys = np.arange(self.rank, cfg.num_keys, self.world, dtype=np.int64)
        for i in range(0, len(my_keys), chunk):
            ks = my_keys[i:i + chunk]
            vals = torch.zeros(len(ks), cfg.row, dtype=torch.float32, device=self.dev)
            # syn0 random-uniform (reference: (rand-0.5)/dim), syn1 zero
            mask = torch.from_numpy((ks % 2 == 0)).to(self.dev)
            vals[:, :cfg.dim] = (torch.rand(len(ks), cfg.dim, device=self.dev) - 0.5) / cfg.dim
            vals[~mask, :cfg.dim] = 0.0
            vals[:, cfg.dim:] = cfg.accum_init
            self.worker.set(ks, vals)
        self.worker.wait_sync()
        self.worker.barrier()

    def set_vocab_counts(self, counts, sample: float = 1e-3):
        """Frequent-word subsampling (reference word2vec.cc 'sample'
        parameter): P(keep w) = min(1, sqrt(t/f) + t/f)."""
        f = np.asarray(counts, dtype=np.float64)
        f = f / f.sum()
        with np.errstate(divide="ignore"):
            keep = np.sqrt(sample / f) + sample / f
        self._keep_prob = np.minimum(1.0, keep)

    def pairs_from_sentences(self, sentences):
        """(center, context) pairs with the reference's shrinking window,
        after frequent-word subsampling when counts are set."""
        if self._keep_prob is not None:
            sentences = [s[self.rng.random(len(s)) < self._keep_prob[s]] for s in sentences]
            sentences = [s for s in sentences if len(s) >= 2]
        ctr, ctx = [], []
        for sent in sentences:
            L = len(sent)
            for i, w in enumerate(sent):
                b = self.rng.integers(1, self.cfg.window + 1)
                for j in range(max(0, i - b), min(L, i + b + 1)):
                    if j != i:
                        ctr.append(w)
                        ctx.append(sent[j])
        return np.array(ctr, dtype=np.int64), np.array(ctx, dtype=np.int64)

    def signal_intent(self, sentences, start, end=0):
        words = np.unique(np.concatenate([np.asarray(s, dtype=np.int64) for s in sentences]))
        self.worker.intent(np.concatenate([syn0(words), syn1(words)]), start, end)

    def train_pairs(self, ctr_words, ctx_words, sync_loss=False, neg_words=None):
        cfg = self.cfg
        w = self.worker
        B = len(ctr_words)
        if neg_words is None:
            if self.server.sampling is not None:
                sid = w.prepare_sample(B * cfg.negative, w.current_clock(),
                                       w.current_clock() + 2)
                neg_words = self.server.sampling.pull(w, sid, B * cfg.negative)
                w.finish_sample(sid)
            else:
                neg_words = self.rng.integers(0, cfg.vocab_size, size=B * cfg.negative)

        k_ctr, k_ctx, k_neg = syn0(ctr_words), syn1(ctx_words), syn1(neg_words)
        all_keys = np.concatenate([k_ctr, k_ctx, k_neg])
        row = cfg.row
        all_v = torch.empty(len(all_keys) * row, dtype=torch.float32, device=self.dev)
        w.wait(w.pull(all_keys, all_v, async_=True))
        c_v = all_v[:B * row].view(B, row)
        x_v = all_v[B * row:2 * B * row].view(B, row)
        n_v = all_v[2 * B * row:].view(B * cfg.negative, row)
        all_d = torch.empty_like(all_v)
        dc = all_d[:B * row].view(B, row)
        dx = all_d[B * row:2 * B * row].view(B, row)
        dn = all_d[2 * B * row:].view(B * cfg.negative, row)
        loss = torch.empty(B, dtype=torch.float32, device=self.dev)
        _C.w2v_sgns_step(c_v, x_v, n_v, dc, dx, dn, loss, cfg.negative, cfg.dim,
                         cfg.lr, cfg.eps)
        pt = w.push(all_keys, all_d, async_=True)
        if pt != -1:
            self._pending.append(pt)
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        return float(loss.mean().item()) if sync_loss else loss

    def train_pairs_fused(self, ctr_words, ctx_words, sync_loss=False,
                          force_general=False):
        """Fused slab-direct SGNS step. world==1 + identity layout uses
        the zero-host-work kernel (Server.w2v_step_fused); otherwise the
        general offsets path runs the fused kernel on all-local pairs
        and routes the remote remainder through the classic path."""
        cfg = self.cfg
        w = self.worker
        raw = self.server.raw
        if self.dev.type != "cuda" and not force_general:
            return self.train_pairs(ctr_words, ctx_words, sync_loss=sync_loss)
        B = len(ctr_words)
        if self.server.sampling is not None:
            sid = w.prepare_sample(B * cfg.negative, w.current_clock(), w.current_clock() + 2)
            neg_words = self.server.sampling.pull(w, sid, B * cfg.negative)
            w.finish_sample(sid)
        else:
            neg_words = self.rng.integers(0, cfg.vocab_size, size=B * cfg.negative)
        neg_words = np.ascontiguousarray(neg_words, dtype=np.int64)
        if self.world == 1 and not force_general and raw.layout_identity():
            loss = raw.w2v_step_fused(
                torch.from_numpy(syn0(ctr_words)), torch.from_numpy(syn1(ctx_words)),
                torch.from_numpy(syn1(neg_words)), cfg.negative, cfg.dim, cfg.lr, cfg.eps)
            return float(loss.mean().item()) if sync_loss else loss
        # defer first-time misses one step (intent usually localizes them
        # within a round); a second miss goes classic immediately
        ctr_words = np.asarray(ctr_words, dtype=np.int64)
        ctx_words = np.asarray(ctx_words, dtype=np.int64)
        negs2 = neg_words.reshape(B, cfg.negative)
        n_def = 0
        if self._deferred is not None:
            dc, dx, dn = self._deferred
            self._deferred = None
            n_def = len(dc)
            ctr_words = np.concatenate([dc, ctr_words])
            ctx_words = np.concatenate([dx, ctx_words])
            negs2 = np.concatenate([dn, negs2])
        loss, missed = raw.w2v_step_fused_general(
            torch.from_numpy(syn0(ctr_words)), torch.from_numpy(syn1(ctx_words)),
            torch.from_numpy(syn1(np.ascontiguousarray(negs2.reshape(-1)))),
            cfg.negative, cfg.dim, cfg.lr, cfg.eps)
        if missed.numel():
            midx = missed.numpy()
            old = midx[midx < n_def]
            fresh = midx[midx >= n_def]
            if len(fresh):
                self._deferred = (ctr_words[fresh], ctx_words[fresh], negs2[fresh])
            if len(old):
                mloss = self.train_pairs(ctr_words[old], ctx_words[old], sync_loss=False,
                                         neg_words=negs2[old].reshape(-1))
                if not torch.is_tensor(mloss):
                    mloss = torch.tensor([mloss])
                loss = torch.cat([loss, mloss.to(loss.device)])
        if loss.numel() == 0:
            return 0.0 if sync_loss else loss
        return float(loss.mean().item()) if sync_loss else loss

    def drain(self):
        if self._deferred is not None:
            dc, dx, dn = self._deferred
            self._deferred = None
            self.train_pairs(dc, dx, sync_loss=False, neg_words=dn.reshape(-1))
        for t in self._pending:
            self.worker.wait(t)
        self._pending.clear()

    def export_text(self, path: str, max_words: int = None, chunk: int = 65536):
        """word2vec text format: '<vocab> <dim>' header then word vectors
        (reference word2vec.cc:367-416)."""
        self.drain()
        self.worker.wait_sync()
        self.worker.barrier()
        if self.rank == 0:
            n = min(self.cfg.vocab_size, max_words or self.cfg.vocab_size)
            with open(path, "w") as f:
                f.write(f"{n} {self.cfg.dim}\n")
                for i in range(0, n, chunk):
                    ks = syn0(np.arange(i, min(i + chunk, n)))
                    buf = np.zeros((len(ks), self.cfg.row), dtype=np.float32)
                    self.worker.pull(ks, buf)
                    for j, wv in enumerate(buf[:, :self.cfg.dim]):
                        name = self.words[i + j] if self.words else f"w{i+j}"
                        f.write(f"{name} " + " ".join(f"{x:.6f}" for x in wv) + "\n")
        self.worker.barrier()

    def export_binary(self, path: str, max_words: int = None, chunk: int = 65536):
        """Classic word2vec BINARY format (reference word2vec.cc:367-380)."""
        from .data_io import export_word2vec_binary

        self.drain()
        self.worker.wait_sync()
        self.worker.barrier()
        if self.rank == 0:
            n = min(self.cfg.vocab_size, max_words or self.cfg.vocab_size)
            vecs = np.zeros((n, self.cfg.dim), dtype=np.float32)
            for i in range(0, n, chunk):
                ks = syn0(np.arange(i, min(i + chunk, n)))
                buf = np.zeros((len(ks), self.cfg.row), dtype=np.float32)
                self.worker.pull(ks, buf)
                vecs[i:i + len(ks)] = buf[:, :self.cfg.dim]
            names = self.words[:n] if self.words else [f"w{i}" for i in range(n)]
            export_word2vec_binary(path, names, vecs)
        self.worker.barrier()


def make_synthetic_sentences(n_sentences, vocab_size, mean_len=12, zipf_a=1.2, seed=0):
    """Zipf-distributed synthetic corpus (no network for real data)."""
    rng = np.random.default_rng(seed)
    out = []
    for _ in range(n_sentences):
        L = max(2, int(rng.poisson(mean_len)))
        words = rng.zipf(zipf_a, size=L)
        out.append(np.minimum(words - 1, vocab_size - 1).astype(np.int64))
    return out


def main():
    """CLI (rebuild of reference apps/word2vec.cc): SGNS on a synthetic
    corpus with subsampling + unigram negatives; exports embeddings."""
    import argparse
    import time

    import adapm_amd as _a

    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=100_000)
    ap.add_argument("--dim", type=int, default=100)
    ap.add_argument("--window", type=int, default=5)
    ap.add_argument("--negative", type=int, default=5)
    ap.add_argument("--sentences", type=int, default=20_000)
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--batch-pairs", type=int, default=16384)
    ap.add_argument("--output", type=str, default="")
    ap.add_argument("--binary-output", type=str, default="",
                    help="also write the classic word2vec binary format")
    ap.add_argument("--corpus", type=str, default="",
                    help="train on a real text corpus (vocabulary built from the "
                         "file, reference LearnVocabFromTrainFile); default: synthetic")
    ap.add_argument("--min-count", type=int, default=5)
    ap.add_argument("--device", type=str, default=None)
    a = ap.parse_args()

    words = None
    if a.corpus:
        from .data_io import build_vocab, read_sentences

        words, wcounts, word2id = build_vocab(a.corpus, min_count=a.min_count,
                                              max_vocab=a.vocab)
        a.vocab = len(words)
        counts = wcounts.astype(np.float64)
    else:
        counts = (1.0 / np.arange(1, a.vocab + 1)) ** 0.75 * 1e9

    cfg = W2VConfig(vocab_size=a.vocab, dim=a.dim, window=a.window, negative=a.negative,
                    batch_pairs=a.batch_pairs)
    _a.setup(num_keys=cfg.num_keys, num_threads=1, device=a.device)
    server = _a.Server(cfg.row)
    server.enable_sampling_support("local", True, "unigram", 0, a.vocab, counts=counts)
    worker = _a.Worker(0, server)
    model = Word2Vec(cfg, server, worker)
    model.set_vocab_counts(counts)
    model.words = words
    model.init_embeddings()
    rank = server.my_rank()
    world = server.rt.world
    if a.corpus:
        sents = [s for i, s in enumerate(read_sentences(a.corpus, word2id))
                 if i % world == rank]  # sentence partition (reference file-offset split)
    else:
        sents = make_synthetic_sentences(a.sentences // world, a.vocab, seed=rank)
    for ep in range(a.epochs):
        t0 = time.time()
        ctr, ctx = model.pairs_from_sentences(sents)
        losses = []
        for i in range(0, len(ctr), a.batch_pairs):
            model.signal_intent([np.unique(ctr[i:i + a.batch_pairs])],
                                worker.current_clock() + 1, worker.current_clock() + 3)
            losses.append(model.train_pairs(ctr[i:i + a.batch_pairs],
                                            ctx[i:i + a.batch_pairs], sync_loss=True))
            worker.advance_clock()
        model.drain()
        total = worker.allreduce(float(np.mean(losses)))
        if rank == 0:
            print(f"[w2v] epoch {ep}: loss {total / world:.4f} "
                  f"({len(ctr)} pairs, {time.time()-t0:.1f}s)")
    if a.output:
        model.export_text(a.output, max_words=min(a.vocab, 10000))
        if rank == 0:
            print(f"[w2v] embeddings -> {a.output}")
    if a.binary_output:
        model.export_binary(a.binary_output, max_words=min(a.vocab, 10000))
        if rank == 0:
            print(f"[w2v] binary embeddings -> {a.binary_output}")
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
