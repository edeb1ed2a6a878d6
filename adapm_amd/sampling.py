"""Sampling access support — the four schemes of the reference
(reference include/ps/sampling.h: Naive, Preloc, Pool, Local), plus the
uniform / log-uniform distributions of the reference bindings
(bindings/bindings.cc:65-77) and a unigram**0.75 extension for word2vec.

Keys are drawn host-side (keys route on the host anyway); the Local
scheme's locality scan runs in C++ over the store's presence metadata
(Server.scan_local)."""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

import numpy as np
import torch


class Distribution:
    def draw(self, n: int) -> np.ndarray:  # int64 keys
        raise NotImplementedError


class Uniform(Distribution):
    def __init__(self, lo: int, hi: int, seed: int):
        self.lo, self.hi = lo, hi
        self.rng = np.random.default_rng(seed)

    def draw(self, n):
        return self.rng.integers(self.lo, self.hi, size=n, dtype=np.int64)


class LogUniform(Distribution):
    """key = floor(exp(u * log(hi-lo+1)) + lo - 1), matching the reference
    binding's LogUniformSampling (bindings.cc:72-77)."""

    def __init__(self, lo: int, hi: int, seed: int):
        self.lo, self.hi = lo, hi
        self.rng = np.random.default_rng(seed)

    def draw(self, n):
        u = self.rng.random(n)
        k = np.exp(u * np.log(self.hi - self.lo + 1)) + self.lo - 1
        return np.minimum(k.astype(np.int64), self.hi - 1)


class Unigram(Distribution):
    """unigram^power sampling (word2vec negative sampling, reference
    apps/word2vec.cc:125-146) via an O(1)-per-draw ALIAS TABLE. On GPU
    stores the draw is a HIP kernel (_C.alias_draw) with the table
    resident on-device; on CPU it is the native C++ loop."""

    def __init__(self, counts: np.ndarray, key_of: Optional[np.ndarray], power: float, seed: int,
                 device=None):
        p = counts.astype(np.float64) ** power
        p = p / p.sum()
        self.prob, self.alias = self._build_alias(p)
        self.key_of = key_of  # optional map index->key
        self.seed = np.random.default_rng(seed).integers(1, 2 ** 62)
        self._calls = 0
        self.device = device
        if device is not None and str(device).startswith("cuda"):
            self.prob = self.prob.to(device)
            self.alias = self.alias.to(device)

    @staticmethod
    def _build_alias(p: np.ndarray):
        import torch as _t

        n = len(p)
        prob = (p * n).astype(np.float64)
        alias = np.zeros(n, dtype=np.int32)
        small = list(np.where(prob < 1.0)[0][::-1])
        large = list(np.where(prob >= 1.0)[0][::-1])
        while small and large:
            s, l = small.pop(), large.pop()
            alias[s] = l
            prob[l] = prob[l] - (1.0 - prob[s])
            (small if prob[l] < 1.0 else large).append(l)
        return (_t.from_numpy(prob.clip(0, 1).astype(np.float32)),
                _t.from_numpy(alias))

    def draw(self, n):
        from . import _C

        self._calls += 1
        out = _C.alias_draw(self.prob, self.alias, int(self.seed + self._calls), n)
        keys = out.cpu().numpy() if out.is_cuda else out.numpy()
        return keys if self.key_of is None else self.key_of[keys]


@dataclass
class _Sample:
    keys: Optional[np.ndarray]  # pre-drawn (naive/preloc/pool); None for local
    remaining: int
    used: set = field(default_factory=set)
    cursor: int = 0


class SamplingManager:
    """Per-rank sampling state shared by all workers (reference sampling.h
    keeps it on the server; sample ids are (wid, counter))."""

    def __init__(self, server, scheme: str, with_replacement: bool, dist_: Distribution,
                 lo: int, hi: int, pool_size: int = 5_000, reuse_factor: int = 4):
        self.POOL_SIZE = pool_size    # reference --sampling.pool_size
        self.POOL_REUSE = reuse_factor  # reference --sampling.reuse_factor
        self.server = server
        self.scheme = scheme
        self.with_replacement = with_replacement
        self.dist = dist_
        self.lo, self.hi = lo, hi
        self.lock = threading.Lock()
        self.samples: Dict[int, _Sample] = {}
        self.next_id = 1
        self.pool: np.ndarray = np.empty(0, dtype=np.int64)
        self.pool_uses = 0

    # ------------------------------------------------ prepare / pull

    def prepare(self, worker, K: int, start: int, end: int) -> int:
        with self.lock:
            sid = self.next_id
            self.next_id += 1
        if self.scheme in ("naive", "preloc", "pool"):
            if self.scheme == "pool":
                keys = self._from_pool(K)
            else:
                keys = self._draw(K)
            self.samples[sid] = _Sample(keys=keys, remaining=K)
            if self.scheme == "preloc" and self.server.world() > 1:
                worker.intent(torch.from_numpy(keys), start, end)
        else:  # local
            self.samples[sid] = _Sample(keys=None, remaining=K)
        return sid

    def pull(self, worker, sid: int, n: int):
        """Choose n concrete keys for sample `sid`. Returns np.int64 keys;
        the caller pulls their values (reference pull_sample)."""
        s = self.samples[sid]
        if s.remaining < n:
            raise ValueError(f"sample {sid}: requested {n} > remaining {s.remaining}")
        if self.scheme == "local":
            keys = self._local_draw(n, s)
        else:
            keys = s.keys[s.cursor:s.cursor + n]
            s.cursor += n
        s.remaining -= n
        return keys

    def finish(self, sid: int):
        self.samples.pop(sid, None)

    # ------------------------------------------------ internals

    def _draw(self, n) -> np.ndarray:
        if self.with_replacement:
            return self.dist.draw(n)
        out = []
        seen = set()
        while len(out) < n:
            cand = self.dist.draw(max(2 * (n - len(out)), 16))
            for c in cand:
                c = int(c)
                if c not in seen:
                    seen.add(c)
                    out.append(c)
                    if len(out) == n:
                        break
        return np.array(out, dtype=np.int64)

    def _from_pool(self, n) -> np.ndarray:
        with self.lock:
            if len(self.pool) < n or self.pool_uses >= self.POOL_REUSE * self.POOL_SIZE:
                self.pool = self.dist.draw(max(self.POOL_SIZE, n))
                self.pool_uses = 0
            idx = np.random.randint(0, len(self.pool), size=n)
            self.pool_uses += n
            return self.pool[idx].copy()

    def _local_draw(self, n, s: _Sample) -> np.ndarray:
        """Draw candidates, substitute each with the next locally-present
        key (reference sampling.h Local scheme, 366-525)."""
        cands = self.dist.draw(n)
        if self.server.world() == 1:
            keys = cands  # single rank: every key is local, no scan needed
        else:
            # snap each candidate to the nearest key this rank OWNS BY
            # CONSTRUCTION (manager = key % world): the scan below then
            # hits on the first probe for everything that has not been
            # relocated away. This is the NuPS "local" substitution —
            # distribution preserved approximately (reference sampling.h
            # Local scheme does the same kind of substitution).
            w, r = self.server.world(), self.server.rank()
            cands = cands - (cands % w) + r
            cands = np.where(cands >= self.hi, cands - w, cands)
            cands = np.where(cands < self.lo, cands + w, cands)
            keys, _checks = self.server.scan_local(torch.from_numpy(cands), self.lo, self.hi)
            keys = keys.numpy()
        if not self.with_replacement:
            out = []
            attempts = 0
            for k in keys:
                kk = int(k)
                while kk in s.used and attempts < 100:
                    kk2 = int(self.dist.draw(1)[0])
                    kk = int(self.server.scan_local(
                        torch.tensor([kk2], dtype=torch.int64), self.lo, self.hi)[0][0])
                    attempts += 1
                s.used.add(kk)
                out.append(kk)
            keys = np.array(out, dtype=np.int64)
        return keys


def make_distribution(name: str, lo: int, hi: int, seed: int, counts=None,
                      power: float = 0.75, device=None) -> Distribution:
    if name == "uniform":
        return Uniform(lo, hi, seed)
    if name == "log-uniform":
        return LogUniform(lo, hi, seed)
    if name == "unigram":
        if counts is None:
            raise ValueError("unigram distribution needs counts")
        return Unigram(np.asarray(counts), None, power, seed, device=device)
    raise ValueError(f"unknown sampling distribution '{name}'")
