"""CTR wide-and-deep with a giant embedding table on the parameter
manager (BASELINE.json config 5: 100M x 64 table, PyTorch bindings path).

The embedding table lives in the KV store (value = [emb(dim) |
AdaGrad accum(dim)]); the dense tower is a small plain-PyTorch MLP
optimized locally per rank. Per batch: pull the batch's feature
embeddings into a device tensor with requires_grad, autograd through the
dense tower, then AdaGrad-transform the embedding gradient and push it —
the same pulled-tensor-autograd pattern the reference's torch bindings
enable (reference bindings/example.py)."""
from __future__ import annotations

import dataclasses

import numpy as np
import torch


@dataclasses.dataclass
class CTRConfig:
    num_features: int = 1_000_000   # embedding-table keys (north star: 100M)
    dim: int = 64
    fields: int = 16                # categorical fields per example
    hidden: int = 256
    batch_size: int = 8192
    lr: float = 0.01
    eps: float = 1e-6
    dense_lr: float = 1e-3
    lookahead: int = 2
    seed: int = 5

    @property
    def row(self):
        return 2 * self.dim


class WideAndDeep:
    def __init__(self, cfg: CTRConfig, server, worker):
        self.cfg = cfg
        self.server = server
        self.worker = worker
        self.dev = server.rt.device
        self.rank = server.rt.rank
        self.world = server.rt.world
        self.rng = np.random.default_rng(cfg.seed + self.rank)
        torch.manual_seed(cfg.seed)
        self.net = torch.nn.Sequential(
            torch.nn.Linear(cfg.fields * cfg.dim, cfg.hidden),
            torch.nn.ReLU(),
            torch.nn.Linear(cfg.hidden, 1),
        ).to(self.dev)
        self.opt = torch.optim.Adam(self.net.parameters(), lr=cfg.dense_lr)
        self._pending = []

    def init_embeddings(self, scale=0.01):
        cfg = self.cfg
        chunk = max(1, 2 ** 25 // cfg.row)
        my_keys = np.arange(self.rank, cfg.num_features, self.world, dtype=np.int64)
        for i in range(0, len(my_keys), chunk):
            ks = my_keys[i:i + chunk]
            vals = torch.zeros(len(ks), cfg.row, dtype=torch.float32, device=self.dev)
            vals[:, :cfg.dim].normal_(0, scale)
            self.worker.set(ks, vals)
        self.worker.wait_sync()
        self.worker.barrier()

    def signal_intent(self, feats, start, end=0):
        self.worker.intent(np.unique(feats.reshape(-1)), start, end)

    def train_batch(self, feats: np.ndarray, labels: np.ndarray, sync_loss=True):
        """feats: [B, fields] int64 feature ids; labels: [B] {0,1}.

        Keys are dedup'd per batch: the store sees each distinct key once
        (Zipf batches repeat hot ids thousands of times), autograd
        aggregates the per-occurrence gradients on-device via the
        index_select backward (a scatter-add in HBM), and AdaGrad then
        transforms the aggregated gradient — one pull row + one push row
        per distinct key."""
        cfg = self.cfg
        w = self.worker
        B, F = feats.shape
        keys = feats.reshape(-1).astype(np.int64)
        if self.dev.type == "cuda":
            keys_u_t, inv = torch.unique(torch.from_numpy(keys).to(self.dev),
                                         return_inverse=True)
            keys_u = keys_u_t.cpu().numpy()
        else:
            keys_u, inv = np.unique(keys, return_inverse=True)
            inv = torch.from_numpy(inv)
        U = len(keys_u)
        rows = torch.empty(U, cfg.row, dtype=torch.float32, device=self.dev)
        w.wait(w.pull(keys_u, rows, async_=True))
        emb = rows[:, :cfg.dim].detach().clone().requires_grad_(True)
        accum = rows[:, cfg.dim:]

        x = emb.index_select(0, inv.to(self.dev)).view(B, F * cfg.dim)
        logits = self.net(x).squeeze(1)
        y = torch.as_tensor(labels.astype(np.float32), device=self.dev)
        loss = torch.nn.functional.binary_cross_entropy_with_logits(logits, y)
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        self.opt.step()

        # AdaGrad-transform the aggregated embedding grads into a
        # push-ready delta (one row per distinct key)
        g = emb.grad
        delta = torch.empty_like(rows)
        g2 = g * g
        delta[:, :cfg.dim] = -cfg.lr * g / torch.sqrt(accum + g2 + cfg.eps)
        delta[:, cfg.dim:] = g2
        pt = w.push(keys_u, delta, async_=True)
        if pt != -1:
            self._pending.append(pt)
        while len(self._pending) > 64:
            w.wait(self._pending.pop(0))
        return float(loss.item()) if sync_loss else loss

    def drain(self):
        for t in self._pending:
            self.worker.wait(t)
        self._pending.clear()


def make_synthetic_ctr(n, num_features, fields, zipf_a=1.1, seed=0):
    rng = np.random.default_rng(seed)
    feats = np.minimum(rng.zipf(zipf_a, size=(n, fields)) - 1, num_features - 1).astype(np.int64)
    # a learnable rule: label depends on a hash of the first field
    labels = ((feats[:, 0] * 2654435761) % 97 < 48).astype(np.int64)
    return feats, labels
