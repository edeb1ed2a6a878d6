"""adapm_amd — MI355X-native adaptive parameter manager.

A from-scratch rebuild of the capabilities of AdaPM (alexrenz/AdaPM,
mounted read-only at /root/reference) for AMD Instinct MI355X:

  - key -> dense-vector parameter store resident in HBM3E (float32 slab,
    HIP gather/scatter kernels for the hot paths),
  - one process per GPU; torch.distributed over RCCL/xGMI replaces the
    reference's ZeroMQ Van; remote ops + the replication/relocation
    protocol ride per-channel batched all-to-all-v sync rounds,
  - intent-driven parameter management: Intent(keys, start, end) +
    advanceClock() trigger relocation or replication ahead of access,
  - sampling access (PrepareSample/PullSample, 4 schemes).

Public surface mirrors the reference PyTorch bindings
(reference bindings/bindings.cc): setup, Server, Worker (pull/push/set/
intent/advance_clock/prepare_sample/pull_sample/wait/waitall/wait_sync/
barrier/...), with torch.Tensor and numpy overloads.
"""
from __future__ import annotations

import threading
from typing import Optional

import numpy as np
import torch

try:
    from . import _C
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "adapm_amd native extension not built — run `python build.py` at the repo root"
    ) from e

from . import runtime as _rt
from .sampling import SamplingManager, make_distribution
from .sync import SyncManager

TECH_ALL = 0
TECH_REPLICATION_ONLY = 1
TECH_RELOCATION_ONLY = 2

_SETUP = {}


def setup(num_keys: int, num_threads: int, use_techniques: str = "", num_channels: int = -1,
          device: Optional[str] = None, max_sync_per_sec: float = 1000.0,
          sync_threshold: float = 0.0,
          time_intent_actions: bool = True, capacity_factor: float = 2.0,
          location_caches: bool = True, locality_stats: bool = False,
          trace_keys=None, stats_out: Optional[str] = None,
          device_cap_gb: float = 0.0, host_spill_gb: float = 0.0):
    """Global configuration (reference bindings.cc:18-31 `setup`)."""
    tech = TECH_ALL
    t = use_techniques.strip().lower()
    if t in ("replication_only", "replication"):
        tech = TECH_REPLICATION_ONLY
    elif t in ("relocation_only", "relocation"):
        tech = TECH_RELOCATION_ONLY
    elif t not in ("", "all"):
        raise ValueError(f"unknown management techniques '{use_techniques}'")
    if num_channels == -1:
        num_channels = 2
    _SETUP.update(dict(num_keys=num_keys, num_threads=num_threads, techniques=tech,
                       num_channels=num_channels, device=device,
                       max_sync_per_sec=max_sync_per_sec,
                       sync_threshold=sync_threshold,
                       time_intent_actions=time_intent_actions,
                       capacity_factor=capacity_factor,
                       location_caches=location_caches, locality_stats=locality_stats,
                       trace_keys=trace_keys, stats_out=stats_out,
                       device_cap_gb=device_cap_gb, host_spill_gb=host_spill_gb))


def scheduler(num_keys: int = 0, num_threads: int = 0):
    """Compat shim: the reference needs a scheduler process for rendezvous
    (reference ps.h:38-42); here rendezvous is the torch.distributed
    TCPStore, so there is no scheduler role. No-op."""
    return None


def my_rank() -> int:
    return _rt.get_runtime().rank


class Server:
    """Per-process server: owns the HBM store, directory, sync threads and
    sampling (reference ColoKVServer, coloc_kv_server.h)."""

    def __init__(self, value_lengths, num_keys: Optional[int] = None):
        if not _SETUP:
            raise RuntimeError("call adapm_amd.setup(...) before creating a Server")
        cfg = _SETUP
        rt = _rt.init_runtime(num_channels=cfg["num_channels"], device=cfg["device"])
        self.rt = rt

        if isinstance(value_lengths, int):
            lens = torch.tensor([value_lengths], dtype=torch.int64)
        elif isinstance(value_lengths, np.ndarray):
            lens = torch.from_numpy(value_lengths.astype(np.int64))
        else:
            lens = value_lengths.to(torch.int64)
        nk = num_keys if num_keys is not None else cfg["num_keys"]
        if lens.numel() not in (1, nk):
            raise ValueError("value_lengths must be a scalar or have num_keys entries")

        if rt.is_cuda and not _C.hip_available():
            raise RuntimeError("CUDA device requested but HIP extension reports no device")

        self._s = _C.Server(
            num_keys=nk, value_lengths=lens, rank=rt.rank, world=rt.world,
            num_channels=cfg["num_channels"], num_workers=cfg["num_threads"],
            device=str(rt.device), capacity_factor=cfg["capacity_factor"],
            techniques=cfg["techniques"], location_caches=cfg["location_caches"],
            device_cap_floats=int(cfg.get("device_cap_gb", 0) * (1 << 30) / 4),
            host_spill_floats=int(cfg.get("host_spill_gb", 0) * (1 << 30) / 4),
            sync_threshold=cfg.get("sync_threshold", 0.0),
        )
        if cfg.get("locality_stats"):
            self._s.enable_locality_stats()
        if cfg.get("trace_keys") is not None:
            tk = cfg["trace_keys"]
            t = torch.tensor([-1], dtype=torch.int64) if tk == "all" else _to_key_tensor(tk)
            self._s.enable_key_trace(t)
        self._stats_out = cfg.get("stats_out")
        self.sampling: Optional[SamplingManager] = None
        self._sync = SyncManager(self._s, rt, max_per_sec=cfg["max_sync_per_sec"],
                                 time_intent_actions=cfg["time_intent_actions"])
        self._sync.start()
        self._local_barrier = threading.Barrier(max(1, cfg["num_threads"]))
        self._num_workers = max(1, cfg["num_threads"])
        self._shut = False

    # ---- reference Server API (bindings.cc:88-146)

    def enable_sampling_support(self, scheme: str, with_replacement: bool, distribution: str,
                                min: int, max: int, counts=None, power: float = 0.75,
                                pool_size: int = 5_000, reuse_factor: int = 4):
        d = make_distribution(distribution, min, max, seed=self.rt.rank, counts=counts,
                              power=power, device=self.rt.device)
        self.sampling = SamplingManager(self._s, scheme, with_replacement, d, min, max,
                                        pool_size=pool_size, reuse_factor=reuse_factor)

    def barrier(self):
        if self.rt.world > 1:
            with self.rt.worker_group_lock:
                torch.distributed.barrier(group=self.rt.worker_group)

    def my_rank(self) -> int:
        return self.rt.rank

    def shutdown(self):
        if self._shut:
            return
        self._shut = True
        self._sync.request_stop()
        self._sync.join()
        self._final_report()
        _rt.shutdown_runtime()

    def _final_report(self):
        """End-of-run locality summary (reference coloc_kv_server.h:147-157)
        + optional TSV dumps (--sys.stats.out equivalent)."""
        import os
        import sys

        st = self.stats()
        if os.environ.get("ADAPM_VERBOSE", "0") != "0":
            pl = st["pull_local"] / max(1, st["pull_keys"])
            ph = st["push_local"] / max(1, st["push_keys"])
            rr, rp = st["replica_records"], st["replica_payloads"]
            pct = 100.0 * rp / rr if rr else 0.0
            hh = st["hop_hist"]
            served = sum(hh)
            mean_hops = (sum(i * h for i, h in enumerate(hh)) / served) if served else 0.0
            print(f"[adapm rank {self.rt.rank}] pulls: {st['pull_keys']} ({pl:.1%} local, "
                  f"{st['pull_replica']} from replicas); pushes: {st['push_keys']} "
                  f"({ph:.1%} local); relocations {st['relocations_out']}/"
                  f"{st['relocations_in']} out/in; replications {st['replications']}; "
                  f"drops {st['replica_drops']}; sync rounds {st['sync_rounds']}; "
                  f"{pct:.1f}% replica records carried payload; remote ops served "
                  f"{served} (mean {mean_hops:.2f} hops, hist {hh}); "
                  f"{st['bytes_sent']/1e6:.1f}/{st['bytes_recv']/1e6:.1f} MB sent/recv",
                  file=sys.stderr, flush=True)
        if self._stats_out:
            os.makedirs(self._stats_out, exist_ok=True)
            r = self.rt.rank
            self._s.dump_locality_stats(os.path.join(self._stats_out,
                                                     f"locality_stats.rank.{r}.tsv"))
            self._s.dump_traces(os.path.join(self._stats_out, f"traces.{r}.tsv"))

    # ---- extras

    def stats(self) -> dict:
        return dict(self._s.stats())

    def wait_sync(self, strong: bool = False):
        """Block until 2 more sync rounds completed on every channel
        (reference WaitSync, coloc_kv_worker.h:517-550).

        strong=True waits instead for a *globally idle* point: on every
        channel, 2 consecutive rounds in which no rank sent or
        self-handled anything. Forwarded deltas can escape the fixed
        2-round window under relocation churn (the reference has the
        same hazard); after a globally idle point every in-flight
        delta/forward/refresh has drained, so all prior pushes are
        visible. Unbounded if other ranks push continuously — use after
        a barrier. (With sync_threshold > 0, deltas below the threshold
        are withheld by design and not covered.)"""
        if self.rt.world <= 1:
            return
        if strong:
            # +2, not +1: the round already in flight at call time was
            # collected before our ops landed and may still complete as
            # "idle"; only one round can be mid-flight (single comm
            # thread), so the second idle event is necessarily observed
            # by a round that saw our work.
            counts = self._s.idle_counts()
            self._s.wait_idle([c + 2 for c in counts])
        else:
            counts = self._s.round_counts()
            self._s.wait_rounds([c + 2 for c in counts])

    @property
    def raw(self):
        return self._s


def _to_key_tensor(keys) -> torch.Tensor:
    if isinstance(keys, np.ndarray):
        t = torch.from_numpy(np.ascontiguousarray(keys, dtype=np.int64))
    elif isinstance(keys, torch.Tensor):
        t = keys.detach()
        if t.device.type != "cpu":
            t = t.cpu()
        t = t.to(torch.int64).contiguous()
    elif isinstance(keys, (list, tuple, range)):
        t = torch.tensor(list(keys), dtype=torch.int64)
    else:
        t = torch.tensor([int(keys)], dtype=torch.int64)
    return t.reshape(-1)


class Worker:
    """Worker-thread API object (reference ColoKVWorker / bindings Worker).

    Ops return a timestamp; -1 means answered entirely locally (for GPU
    tensors: the kernels are enqueued on the current stream). `wait(ts)`
    blocks until remote responses arrived.
    """

    def __init__(self, customer_id: int, server: Server):
        self.server = server
        self.wid = customer_id
        self._s = server._s
        self._ulen = self._s.uniform_len()

    # ---- data ops (torch / numpy overloads, like bindings.cc:160-290)

    def _vals_tensor(self, vals):
        if isinstance(vals, np.ndarray):
            if vals.dtype != np.float32:
                raise TypeError("numpy vals must be float32")
            return torch.from_numpy(vals), vals
        if not isinstance(vals, torch.Tensor):
            raise TypeError("vals must be a torch.Tensor or numpy float32 array")
        if vals.dtype != torch.float32:
            raise TypeError("vals must be float32")
        return vals, None

    def _check_len(self, kt, vt):
        """Always-on size validation (reference bindings.cc:174-186). The
        uniform-length check is O(1) here and raises ValueError; per-key
        length stores are validated by the C++ core on every call."""
        if self._ulen >= 0:
            need = kt.numel() * self._ulen
            if need != vt.numel():
                raise ValueError(f"value array has {vt.numel()} floats, "
                                 f"{kt.numel()} key(s) need {need}")

    def pull(self, keys, vals, async_: bool = False, **kw):
        async_ = kw.get("async", async_)
        kt = _to_key_tensor(keys)
        vt, _ = self._vals_tensor(vals)
        self._check_len(kt, vt)
        ts = self._s.pull(self.wid, kt, vt)
        if ts != -1:
            self.server._sync.kick_event.set()
        if not async_:
            self._s.wait(ts)
        return ts

    def push(self, keys, vals, async_: bool = False, **kw):
        async_ = kw.get("async", async_)
        kt = _to_key_tensor(keys)
        vt, _ = self._vals_tensor(vals)
        self._check_len(kt, vt)
        ts = self._s.push(self.wid, kt, vt, False)
        if ts != -1:
            self.server._sync.kick_event.set()
        if not async_:
            self._s.wait(ts)
        return ts

    def set(self, keys, vals, async_: bool = False, **kw):
        async_ = kw.get("async", async_)
        kt = _to_key_tensor(keys)
        vt, _ = self._vals_tensor(vals)
        self._check_len(kt, vt)
        ts = self._s.push(self.wid, kt, vt, True)
        if ts != -1:
            self.server._sync.kick_event.set()
        if not async_:
            self._s.wait(ts)
        return ts

    def pull_if_local(self, keys, vals) -> bool:
        kt = _to_key_tensor(keys)
        vt, _ = self._vals_tensor(vals)
        return self._s.pull_if_local(kt, vt)

    def is_local(self, key: int) -> bool:
        return self._s.is_local(int(key))

    # ---- intent / clock

    def intent(self, keys, start: int, end: int = 0):
        kt = _to_key_tensor(keys)
        self._s.intent(self.wid, kt, start, end)

    def advance_clock(self):
        self._s.advance_clock(self.wid)

    def current_clock(self) -> int:
        return self._s.current_clock(self.wid)

    # ---- sampling (reference coloc_kv_worker.h:418-442)

    def prepare_sample(self, K: int, start: int = 0, end: int = 0) -> int:
        smp = self.server.sampling
        if smp is None:
            raise RuntimeError("sampling support not enabled on this server")
        return smp.prepare(self, K, start, end)

    def pull_sample(self, sample_id: int, keys, vals, async_: bool = False, **kw):
        async_ = kw.get("async", async_)
        smp = self.server.sampling
        if smp is None:
            raise RuntimeError("sampling support not enabled on this server")
        if isinstance(keys, np.ndarray):
            n = keys.shape[0]
            chosen = smp.pull(self, sample_id, n)
            keys[:] = chosen
        else:
            n = keys.shape[0]
            chosen = smp.pull(self, sample_id, n)
            keys.copy_(torch.from_numpy(chosen.astype(np.int64)))
        return self.pull(chosen, vals, async_=async_)

    def finish_sample(self, sample_id: int):
        smp = self.server.sampling
        if smp is not None:
            smp.finish(sample_id)

    # ---- synchronization

    def wait(self, ts: int):
        self._s.wait(ts)

    def waitall(self):
        self._s.wait_all()

    def is_finished(self, ts: int) -> bool:
        return self._s.is_finished(ts)

    def wait_sync(self, strong: bool = False):
        self.server.wait_sync(strong=strong)

    def wait_replica_sync(self):  # deprecated alias (reference bindings.cc:355)
        import warnings

        warnings.warn("wait_replica_sync() is deprecated; use wait_sync()",
                      DeprecationWarning)
        self.wait_sync()

    def barrier(self):
        """Barrier across ALL worker threads of ALL ranks (reference
        Barrier(kWorkerThreadGroup))."""
        i = self.server._local_barrier.wait()
        if i == 0:
            self.server.barrier()
        self.server._local_barrier.wait()

    def begin_setup(self):
        self.barrier()

    def end_setup(self):
        self.wait_sync()
        self.barrier()

    def finalize(self):
        self.waitall()
        self.wait_sync()
        self.barrier()

    def staggered_push(self, keys, vals, chunk: int = 65536):
        """Memory-bounded bulk push: chunked pushes, each waited before
        the next chunk's values are touched (reference StaggeredPush,
        coloc_kv_worker.h:556-580)."""
        kt = _to_key_tensor(keys)
        vt, _ = self._vals_tensor(vals)
        flat = vt.reshape(vt.shape[0], -1) if vt.dim() > 1 else vt.reshape(kt.numel(), -1)
        prev = -1
        for i in range(0, kt.numel(), chunk):
            ks = kt[i:i + chunk]
            vs = flat[i:i + chunk]
            if prev != -1:
                self._s.wait(prev)
            prev = self._s.push(self.wid, ks.contiguous(), vs.contiguous(), False)
            if prev != -1:
                self.server._sync.kick_event.set()
        self._s.wait(prev)

    # ---- collectives (replaces reference utils.h ps_allreduce)

    def allreduce(self, value, op: str = "sum"):
        """Reduce a scalar or tensor across ranks (loss/eval aggregation;
        replaces reference utils.h ps_allreduce). op: sum | max | min.
        Rides the gloo control-plane group (host TCP): worker threads may
        call this concurrently with sync rounds, and NCCL traffic from
        two threads on one device can deadlock — gloo cannot."""
        rt = self.server.rt
        scalar = not isinstance(value, torch.Tensor)
        t = torch.tensor([float(value)]) if scalar else value
        if rt.world > 1:
            import torch.distributed as dist

            ops = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
                   "min": dist.ReduceOp.MIN}
            td = t.to("cpu")
            with rt.worker_group_lock:
                dist.all_reduce(td, op=ops[op], group=rt.worker_group)
            t = td.to(t.device)
        return float(t.item()) if scalar else t

    # ---- info

    def get_key_size(self, key_id: int = 0) -> int:
        return self._s.get_len(int(key_id))

    @property
    def num_keys(self) -> int:
        return self._s.num_keys()

    @property
    def workerId(self) -> int:
        return self.wid
