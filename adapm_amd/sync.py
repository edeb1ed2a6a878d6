"""Per-channel sync loop: the RCCL-over-xGMI replacement of the reference
SyncManager (reference include/ps/sync_manager.h).

Every channel runs one thread on every rank, in lockstep rounds:

  phase A: server.sync_collect(ch)  -> per-dest {replica deltas, replica
           requests, remote Pull/Push/Set requests, forwards}
           size all-gather  +  batched P2P all-to-all-v
           server.sync_process(ch, src, ...) for each incoming message
  phase B: server.sync_respond(ch)  -> per-dest {refreshes, relocations,
           pull responses, push acks, residence updates}
           size all-gather  +  batched P2P all-to-all-v
           server.sync_apply(ch, src, ...)
  server.sync_finish(ch)

On GPU the P2P exchange is dist.batch_isend_irecv = grouped ncclSend/Recv
over the xGMI point-to-point links (per-destination batching, exactly the
per-destination message builds of the reference, sync_manager.h:305,361 —
but without rings, since xGMI is per-link P2P). On CPU it is gloo over
loopback, which is the multi-process no-GPU test tier.

Rounds are collective, so they run in lockstep on every rank; shutdown is
coordinated through a flag column in the size matrix.
"""
from __future__ import annotations

import math
import os
import threading
import time

import torch
import torch.distributed as dist

_TRACE_KEY = int(os.environ.get("ADAPM_TRACE_KEY", "-1"))


_BULK_CODES = {4: 2, 5: 1, 14: 2}  # code -> extra words per key


def _trace(rank, ch, direction, peer, meta):
    if _TRACE_KEY == -1:
        return
    m = meta.reshape(-1).tolist()
    pos = 0
    while pos + 5 <= len(m):
        code, a, f0, f1, f2 = m[pos:pos + 5]
        pos += 5
        if code in _BULK_CODES:
            nk = a
            keys = m[pos:pos + nk]
            pos += _BULK_CODES[code] * nk
            if _TRACE_KEY == -2 or _TRACE_KEY in keys:
                print(f"[trace r{rank} ch{ch} {direction} peer{peer}] BULK code={code} "
                      f"nk={nk} f0={f0} f1={f1} f2={f2} keys={keys[:8]}...", flush=True)
        elif _TRACE_KEY == -2 or a == _TRACE_KEY:
            print(f"[trace r{rank} ch{ch} {direction} peer{peer}] code={code} key={a} "
                  f"f0={f0} f1={f1} f2={f2}", flush=True)


class ActionTimer:
    """Estimate how many worker clock ticks pass per sync round, to decide
    how far ahead of a worker's clock to act on its intents (reference
    sync_manager.h:62-105 estimate_sync_windows_and_tune)."""

    def __init__(self, alpha: float = 0.1, rounds_ahead: float = 2.0):
        self.alpha = alpha
        self.rounds_ahead = rounds_ahead
        self.rate = 0.0  # smoothed clocks per round (max over workers)
        self._last_clocks = None

    def update(self, clocks):
        if self._last_clocks is None:
            self._last_clocks = list(clocks)
            return self.ahead()
        d = max((c - l) for c, l in zip(clocks, self._last_clocks))
        self._last_clocks = list(clocks)
        self.rate = (1 - self.alpha) * self.rate + self.alpha * max(0, d)
        return self.ahead()

    def ahead(self) -> int:
        # rate*rounds_ahead plus a Poisson-style safety buffer
        r = self.rate * self.rounds_ahead
        return int(math.ceil(r + 3.0 * math.sqrt(r + 1.0) + 8.0))


class SyncManager:
    """Drives the sync loop threads for all channels of one rank."""

    def __init__(self, server, runtime, max_per_sec: float = 1000.0,
                 time_intent_actions: bool = True):
        self.server = server
        self.rt = runtime
        self.min_period = 1.0 / max_per_sec if max_per_sec > 0 else 0.0
        self.time_intent_actions = time_intent_actions
        self.stop_requested = threading.Event()
        self.kick_event = threading.Event()
        self.failed = False
        from collections import defaultdict

        self.phase_totals = defaultdict(float)
        self.threads = []
        self.timer = ActionTimer()
        if not time_intent_actions:
            server.set_intent_ahead(1 << 40)

    def start(self):
        if self.rt.world <= 1:
            return  # single rank: nothing to sync (reference sync_manager.h:454-457)
        for ch in range(self.rt.num_channels):
            t = threading.Thread(target=self._loop, args=(ch,), daemon=True,
                                 name=f"adapm-sync-ch{ch}")
            t.start()
            self.threads.append(t)

    def request_stop(self):
        self.stop_requested.set()
        self.kick_event.set()

    def kick(self):
        """Wake the sync loops early: a worker enqueued remote ops and is
        (or will be) waiting on the round. All ranks under symmetric load
        kick at similar times, so the collective rounds speed up together;
        an early kicker just reaches the size all-gather sooner."""
        self.kick_event.set()

    def join(self):
        for t in self.threads:
            t.join()
        self.threads = []
        import os
        import sys

        if os.environ.get("ADAPM_VERBOSE", "0") != "0" and self.phase_totals.get("rounds"):
            pt = dict(self.phase_totals)
            n = pt.pop("rounds")
            print(f"[adapm sync r{self.rt.rank}] {int(n)} rounds; per-round ms: " +
                  ", ".join(f"{k}={1000*v/n:.2f}" for k, v in sorted(pt.items())),
                  file=sys.stderr, flush=True)

    # ---------------------------------------------------------------- loop

    def _loop(self, ch: int):
        try:
            self._loop_inner(ch)
        except Exception as e:  # transport failure: a peer likely died
            import sys

            print(f"[adapm] rank {self.rt.rank}: sync channel {ch} failed: "
                  f"{type(e).__name__}: {e}", file=sys.stderr, flush=True)
            self.failed = True
            self.server.fail(f"sync channel {ch}: {e}")

    def _loop_inner(self, ch: int):
        rt = self.rt
        group = rt.channel_groups[ch]
        world, rank = rt.world, rt.rank
        dev = rt.device if rt.backend == "nccl" else torch.device("cpu")
        if rt.is_cuda:
            torch.cuda.set_device(rt.device)

        verbose = __import__("os").environ.get("ADAPM_VERBOSE", "0") != "0"
        last_report = time.monotonic()
        rounds_at_report = 0
        n_rounds = 0
        while True:
            t0 = time.monotonic()
            if verbose and ch == 0 and t0 - last_report > 10.0:
                clocks = self.server.worker_clocks()
                print(f"[adapm sync r{rank} ch{ch}] {(n_rounds - rounds_at_report) / (t0 - last_report):.0f} rounds/s, "
                      f"worker clocks {clocks}", flush=True)
                last_report, rounds_at_report = t0, n_rounds
            if ch == 0 and self.time_intent_actions:
                self.server.set_intent_ahead(self.timer.update(self.server.worker_clocks()))

            stop = self.stop_requested.is_set()
            all_stopped, any_work = self._round(ch, group, world, rank, dev, stop)
            self.server.sync_finish(ch)
            n_rounds += 1
            if all_stopped:
                return
            # pacing: kick-able sleep; when NO rank had work this round,
            # back off harder (idle rounds are pure overhead)
            period = self.min_period if any_work else max(self.min_period, 0.005)
            dt = time.monotonic() - t0
            if dt < period:
                self.kick_event.wait(timeout=period - dt)
                self.kick_event.clear()

    def _round(self, ch, group, world, rank, dev, stop_flag):
        t0 = time.perf_counter()
        out_a = self.server.sync_collect(ch)
        t1 = time.perf_counter()
        all_stopped, work_a = self._exchange(ch, group, world, rank, dev, out_a,
                                             self.server.sync_process, stop_flag)
        t2 = time.perf_counter()
        out_b = self.server.sync_respond(ch)
        t3 = time.perf_counter()
        _, work_b = self._exchange(ch, group, world, rank, dev, out_b,
                                   self.server.sync_apply, stop_flag)
        t4 = time.perf_counter()
        pt = self.phase_totals
        pt["collect"] += t1 - t0
        pt["exchange_a"] += t2 - t1
        pt["respond"] += t3 - t2
        pt["exchange_b"] += t4 - t3
        pt["rounds"] += 1
        return all_stopped, (work_a or work_b)

    def _exchange(self, ch, group, world, rank, dev, outgoing, handler, stop_flag) -> bool:
        # size matrix: row = this rank's (n_meta_i64, n_payload_f32) per dest
        # + one stop flag in the last column of dest 0's slot
        sizes = torch.zeros(world, 3, dtype=torch.int64)
        msgs = {}
        for dest, meta, payload in outgoing:
            _trace(rank, ch, "out", dest, meta)
            if dest == rank:
                handler(ch, rank, meta, payload)
                continue
            sizes[dest, 0] = meta.numel()
            sizes[dest, 1] = payload.numel()
            msgs[dest] = (meta, payload)
        sizes[:, 2] = 1 if stop_flag else 0

        sizes_d = sizes.to(dev, non_blocking=False)
        gathered = [torch.zeros_like(sizes_d) for _ in range(world)]
        dist.all_gather(gathered, sizes_d, group=group)
        gathered = [g.cpu() for g in gathered]
        all_stopped = all(int(g[0, 2]) == 1 for g in gathered)
        any_work = any(int(g[:, :2].sum()) > 0 for g in gathered)

        # post sends/recvs (meta then payload per peer; order pairs them)
        p2p = []
        recv_bufs = {}
        for peer in range(world):
            if peer == rank:
                continue
            n_meta = int(gathered[peer][rank, 0])
            n_pay = int(gathered[peer][rank, 1])
            if n_meta > 0 or n_pay > 0:
                rm = torch.empty(n_meta, dtype=torch.int64, device=dev)
                rp = torch.empty(n_pay, dtype=torch.float32, device=dev)
                recv_bufs[peer] = (rm, rp)
                if n_meta:
                    p2p.append(dist.P2POp(dist.irecv, rm, peer, group))
                if n_pay:
                    p2p.append(dist.P2POp(dist.irecv, rp, peer, group))
            if peer in msgs:
                meta, payload = msgs[peer]
                sm = meta.reshape(-1).to(dev)
                sp = payload.to(dev) if payload.device != dev else payload
                if sm.numel():
                    p2p.append(dist.P2POp(dist.isend, sm, peer, group))
                if sp.numel():
                    p2p.append(dist.P2POp(dist.isend, sp, peer, group))
        if p2p:
            reqs = dist.batch_isend_irecv(p2p)
            for r in reqs:
                r.wait()
        # handle incoming in fixed rank order for determinism
        store_dev = self.rt.device
        th0 = time.perf_counter()
        for peer in sorted(recv_bufs):
            rm, rp = recv_bufs[peer]
            meta = rm.cpu()
            _trace(rank, ch, "in ", peer, meta)
            payload = rp if rp.device == store_dev else rp.to(store_dev)
            handler(ch, peer, meta, payload)
        self.phase_totals["handlers"] += time.perf_counter() - th0
        return all_stopped, any_work
