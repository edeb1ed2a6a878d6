"""Matrix factorization on the adaptive parameter manager.

Rebuild of the reference app (reference apps/matrix_factorization.cc) with
its three schedules (:409-579):
  - plain_sgd: per-batch look-ahead intent, fully async
  - dsgd: latin-square block schedule — each worker processes disjoint
    column blocks per subepoch, intent for the NEXT block's column keys,
    barrier per subepoch (reference :409-458, apps/mf/data.h:182-192)
  - columnwise: per-process column partitions with per-column intent
    windows (reference :459-522)

Keys: row i -> i, column j -> num_rows + j; value = [factor(R) |
AdaGrad accum(R)] (reference param_len = 2*rank, apps/mf/update.h:23-79).
Loss: NZSL + L2 via allreduce (reference :99-187).
"""
from __future__ import annotations

import dataclasses

import numpy as np
import torch

from adapm_amd import _C


@dataclasses.dataclass
class MFConfig:
    num_rows: int = 1_000_000
    num_cols: int = 100_000
    rank: int = 128
    lr: float = 0.01
    lam: float = 0.05
    eps: float = 1e-6
    batch_nnz: int = 16384
    lookahead: int = 2
    accum_init: float = 1.0  # AdaGrad accumulator start
    seed: int = 11

    @property
    def num_keys(self):
        return self.num_rows + self.num_cols

    @property
    def row(self):
        return 2 * self.rank


class MF:
    def __init__(self, cfg: MFConfig, server, worker):
        self.cfg = cfg
        self.server = server
        self.worker = worker
        self.dev = server.rt.device
        self.rank_id = server.rt.rank
        self.world = server.rt.world
        self.rng = np.random.default_rng(cfg.seed + self.rank_id)
        self._pending = []
        self._deferred = None  # fused-general: missed nonzeros retried next step

    def col_key(self, j):
        return self.cfg.num_rows + np.asarray(j, dtype=np.int64)

    def init_factors(self, scale=0.1):
        cfg = self.cfg
        chunk = max(1, 2 ** 25 // cfg.row)
        my_keys = np.arange(self.rank_id, cfg.num_keys, self.world, dtype=np.int64)
        for i in range(0, len(my_keys), chunk):
            ks = my_keys[i:i + chunk]
            vals = torch.zeros(len(ks), cfg.row, dtype=torch.float32, device=self.dev)
            vals[:, :cfg.rank].uniform_(0, scale)
            vals[:, cfg.rank:] = cfg.accum_init
            self.worker.set(ks, vals)
        self.worker.wait_sync()
        self.worker.barrier()

    def train_batch(self, rows, cols, ratings, sync_loss=False):
        cfg = self.cfg
        w = self.worker
        B = len(rows)
        k_w = np.asarray(rows, dtype=np.int64)
        k_h = self.col_key(cols)
        all_keys = np.concatenate([k_w, k_h])
        row = cfg.row
        all_v = torch.empty(2 * B * row, dtype=torch.float32, device=self.dev)
        w.wait(w.pull(all_keys, all_v, async_=True))
        w_v = all_v[:B * row].view(B, row)
        h_v = all_v[B * row:].view(B, row)
        x = torch.as_tensor(np.asarray(ratings, dtype=np.float32), device=self.dev)
        all_d = torch.empty_like(all_v)
        dw = all_d[:B * row].view(B, row)
        dh = all_d[B * row:].view(B, row)
        loss = torch.empty(B, dtype=torch.float32, device=self.dev)
        _C.mf_update_step(w_v, h_v, x, dw, dh, loss, cfg.rank, cfg.lr, cfg.lam, cfg.eps)
        pt = w.push(all_keys, all_d, async_=True)
        if pt != -1:
            self._pending.append(pt)
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        return float(loss.mean().item()) if sync_loss else loss

    def train_batch_fused(self, rows, cols, ratings, sync_loss=False,
                          force_general=False):
        """Fused slab-direct MF step. world==1 + identity layout uses the
        zero-host-work kernel (Server.mf_step_fused); otherwise the
        general offsets path runs the fused kernel on all-local nonzeros
        and routes the remote remainder through the classic path."""
        cfg = self.cfg
        raw = self.server.raw
        if self.dev.type != "cuda" and not force_general:
            return self.train_batch(rows, cols, ratings, sync_loss=sync_loss)
        k_w = torch.from_numpy(np.asarray(rows, dtype=np.int64))
        k_h = torch.from_numpy(self.col_key(cols))
        x = torch.from_numpy(np.asarray(ratings, dtype=np.float32))
        if self.world == 1 and not force_general and raw.layout_identity():
            loss = raw.mf_step_fused(k_w, k_h, x, cfg.rank, cfg.lr, cfg.lam, cfg.eps)
            return float(loss.mean().item()) if sync_loss else loss
        # defer first-time misses one step (intent usually localizes
        # them within a round); a second miss goes classic immediately
        rows = np.asarray(rows, dtype=np.int64)
        cols = np.asarray(cols, dtype=np.int64)
        ratings = np.asarray(ratings, dtype=np.float32)
        n_def = 0
        if self._deferred is not None:
            dr, dcs, drt = self._deferred
            self._deferred = None
            n_def = len(dr)
            rows = np.concatenate([dr, rows])
            cols = np.concatenate([dcs, cols])
            ratings = np.concatenate([drt, ratings])
            k_w = torch.from_numpy(rows)
            k_h = torch.from_numpy(self.col_key(cols))
            x = torch.from_numpy(ratings)
        loss, missed = raw.mf_step_fused_general(k_w, k_h, x, cfg.rank, cfg.lr, cfg.lam,
                                                 cfg.eps)
        if missed.numel():
            midx = missed.numpy()
            old_i = midx[midx < n_def]
            fresh = midx[midx >= n_def]
            if len(fresh):
                self._deferred = (rows[fresh], cols[fresh], ratings[fresh])
            if len(old_i):
                mloss = self.train_batch(rows[old_i], cols[old_i], ratings[old_i],
                                         sync_loss=False)
                if not torch.is_tensor(mloss):
                    mloss = torch.tensor([mloss])
                loss = torch.cat([loss, mloss.to(loss.device)])
        if loss.numel() == 0:
            return 0.0 if sync_loss else loss
        return float(loss.mean().item()) if sync_loss else loss

    def drain(self):
        if self._deferred is not None:
            dr, dcs, drt = self._deferred
            self._deferred = None
            self.train_batch(dr, dcs, drt, sync_loss=False)
        for t in self._pending:
            self.worker.wait(t)
        self._pending.clear()

    # -------------------------------------------------- schedules

    def epoch_plain_sgd(self, rows, cols, ratings):
        """Per-batch look-ahead intent, fully async (reference :523-579)."""
        cfg = self.cfg
        n = len(rows)
        order = self.rng.permutation(n)
        losses = []
        nb = (n + cfg.batch_nnz - 1) // cfg.batch_nnz
        for b in range(nb):
            idx = order[b * cfg.batch_nnz:(b + 1) * cfg.batch_nnz]
            if b + cfg.lookahead < nb:
                f = order[(b + cfg.lookahead) * cfg.batch_nnz:(b + cfg.lookahead + 1) * cfg.batch_nnz]
                keys = np.concatenate([np.asarray(rows[f], dtype=np.int64), self.col_key(cols[f])])
                self.worker.intent(keys, self.worker.current_clock() + cfg.lookahead,
                                   self.worker.current_clock() + cfg.lookahead + 2)
            losses.append(self.train_batch(rows[idx], cols[idx], ratings[idx], sync_loss=True))
            self.worker.advance_clock()
        self.drain()
        return float(np.mean(losses))

    def epoch_dsgd(self, rows, cols, ratings):
        """Latin-square block schedule: at subepoch t, worker r owns column
        block (r + t) % world; barrier between subepochs (reference
        :409-458)."""
        cfg = self.cfg
        world, r = self.world, self.rank_id
        blk = (cfg.num_cols + world - 1) // world
        col_block = (cols // blk).astype(np.int64)
        losses = []
        for t in range(world):
            my_block = (r + t) % world
            nxt_block = (r + t + 1) % world
            # intent for next subepoch's column keys
            nxt_cols = np.unique(cols[col_block == nxt_block])
            if len(nxt_cols):
                self.worker.intent(self.col_key(nxt_cols), self.worker.current_clock() + 1,
                                   self.worker.current_clock() + 3)
            sel = np.where(col_block == my_block)[0]
            for b in range(0, len(sel), cfg.batch_nnz):
                idx = sel[b:b + cfg.batch_nnz]
                losses.append(self.train_batch(rows[idx], cols[idx], ratings[idx],
                                               sync_loss=True))
            self.worker.advance_clock()
            self.drain()
            self.worker.barrier()
        return float(np.mean(losses)) if losses else 0.0

    def epoch_columnwise(self, rows, cols, ratings):
        """Static per-process column partition with per-column intent
        windows (reference :459-522)."""
        cfg = self.cfg
        world, r = self.world, self.rank_id
        mine = (cols % world) == r
        order = np.argsort(cols[mine], kind="stable")
        sel = np.where(mine)[0][order]
        losses = []
        for b in range(0, len(sel), cfg.batch_nnz):
            idx = sel[b:b + cfg.batch_nnz]
            cs = np.unique(cols[idx])
            self.worker.intent(self.col_key(cs), self.worker.current_clock(),
                               self.worker.current_clock() + 2)
            losses.append(self.train_batch(rows[idx], cols[idx], ratings[idx], sync_loss=True))
            self.worker.advance_clock()
        self.drain()
        return float(np.mean(losses)) if losses else 0.0

    def test_loss(self, rows, cols, ratings, include_reg: bool = False):
        """NZSL (+ optional L2 regularizer) loss over a sample via the
        dedicated reduction kernel (reference apps/mf/loss.h:49-120),
        aggregated across ranks."""
        B = len(rows)
        cfg = self.cfg
        k_w = np.asarray(rows, dtype=np.int64)
        k_h = self.col_key(cols)
        wv = torch.zeros(B, cfg.row, dtype=torch.float32, device=self.dev)
        hv = torch.zeros(B, cfg.row, dtype=torch.float32, device=self.dev)
        self.worker.pull(k_w, wv)
        self.worker.pull(k_h, hv)
        x = torch.as_tensor(np.asarray(ratings, dtype=np.float32), device=self.dev)
        out2 = _C.mf_loss(wv, hv, x, cfg.rank, cfg.lam)
        se = float(out2[0].item()) + (float(out2[1].item()) if include_reg else 0.0)
        return self.worker.allreduce(se) / self.worker.allreduce(float(B))


def _chunks(n, c):
    for i in range(0, n, c):
        yield i, min(i + c, n)


def save_factors(model: "MF", path: str, chunk: int = 65536):
    """rank-0 full pull of row+column factors (incl. AdaGrad state) ->
    .npz (reference matrix_factorization.cc:256-272 factor export;
    loading them back is the resume path, :238-241)."""
    model.drain()
    model.worker.wait_sync()
    model.worker.barrier()
    if model.rank_id == 0:
        cfg = model.cfg
        out = np.empty((cfg.num_keys, cfg.row), dtype=np.float32)
        for a, b in _chunks(cfg.num_keys, chunk):
            buf = np.zeros((b - a, cfg.row), dtype=np.float32)
            model.worker.pull(np.arange(a, b, dtype=np.int64), buf)
            out[a:b] = buf
        np.savez(path, w=out[:cfg.num_rows], h=out[cfg.num_rows:],
                 rank=cfg.rank, rows=cfg.num_rows, cols=cfg.num_cols)
    model.worker.barrier()


def load_factors(model: "MF", path: str, chunk: int = 65536):
    if model.rank_id == 0:
        data = np.load(path)
        vals = np.concatenate([data["w"], data["h"]])
        for a, b in _chunks(len(vals), chunk):
            model.worker.set(np.arange(a, b, dtype=np.int64),
                             np.ascontiguousarray(vals[a:b]))
    model.worker.wait_sync()
    model.worker.barrier()


def make_synthetic_ratings(n, num_rows, num_cols, rank_true=8, seed=0):
    """Low-rank synthetic ratings so MF can actually fit them."""
    rng = np.random.default_rng(seed)
    rows = rng.integers(0, num_rows, size=n)
    cols = rng.integers(0, num_cols, size=n)
    u = rng.normal(size=(num_rows, rank_true)) / np.sqrt(rank_true)
    v = rng.normal(size=(num_cols, rank_true)) / np.sqrt(rank_true)
    ratings = np.einsum("ij,ij->i", u[rows], v[cols]).astype(np.float32)
    return rows, cols, ratings


def main():
    """CLI (rebuild of reference apps/matrix_factorization.cc): MF on
    synthetic low-rank ratings with a selectable schedule."""
    import argparse
    import time

    import adapm_amd as _a

    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=100_000)
    ap.add_argument("--cols", type=int, default=10_000)
    ap.add_argument("--rank", type=int, default=64)
    ap.add_argument("--nnz", type=int, default=1_000_000)
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--schedule", choices=["plain_sgd", "dsgd", "columnwise"],
                    default="dsgd")
    ap.add_argument("--lr", type=float, default=0.02)
    ap.add_argument("--data", type=str, default="",
                    help="train on a MatrixMarket .mma ratings file (reference "
                         "apps/mf/io.h); default: synthetic low-rank ratings")
    ap.add_argument("--device", type=str, default=None)
    a = ap.parse_args()

    data = None
    if a.data:
        from .data_io import read_matrix_market

        rows, cols, ratings, (m, n) = read_matrix_market(a.data)
        a.rows, a.cols = m, n
        data = (rows, cols, ratings)

    cfg = MFConfig(num_rows=a.rows, num_cols=a.cols, rank=a.rank, lr=a.lr)
    _a.setup(num_keys=cfg.num_keys, num_threads=1, device=a.device)
    server = _a.Server(cfg.row)
    worker = _a.Worker(0, server)
    model = MF(cfg, server, worker)
    model.init_factors()
    rank_id = server.my_rank()
    world = server.rt.world
    if data is not None:
        rows, cols, ratings = data
    else:
        rows, cols, ratings = make_synthetic_ratings(a.nnz, a.rows, a.cols, seed=7)
    mine = rows % world == rank_id  # row partition (reference data split)
    rows, cols, ratings = rows[mine], cols[mine], ratings[mine]
    ep_fn = getattr(model, f"epoch_{a.schedule}")
    for ep in range(a.epochs):
        t0 = time.time()
        tr = ep_fn(rows, cols, ratings)
        te = model.test_loss(rows[:20000], cols[:20000], ratings[:20000])
        if rank_id == 0:
            print(f"[mf/{a.schedule}] epoch {ep}: train {tr:.4f} test {te:.4f} "
                  f"({time.time()-t0:.1f}s)")
    worker.finalize()
    server.shutdown()


if __name__ == "__main__":
    main()
