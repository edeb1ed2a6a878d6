"""The sync engine: RCCL-over-xGMI replacement of the reference
SyncManager (reference include/ps/sync_manager.h).

ONE comm thread per rank drives ALL channels in lockstep "superrounds":

  phase A: for every channel, server.sync_collect(ch) -> per-dest
           {replica deltas, replica requests, remote Pull/Push/Set
           requests, forwards}; ONE size all-gather covering all
           channels + ONE batched P2P all-to-all-v; then
           server.sync_process(ch, src, ...) per incoming message.
  phase B: same with server.sync_respond(ch) -> {refreshes, relocations,
           pull responses, push acks, NACKs, residence updates} and
           server.sync_apply(ch, src, ...).
  server.sync_finish(ch, globally_idle) per channel.

Why one thread (round-1 ran a thread per channel on its own group):
with NCCL, concurrent collectives on multiple communicators sharing one
device deadlock unless every rank issues them in the same global order.
A single comm thread alternating TWO fixed process groups (size
all-gathers on one, batched P2P on the other — separate gloo contexts
avoid collective/P2P interleaving stalls) makes the issuance order
identical on every rank by construction — and merging the channels'
size all-gathers into one cuts per-round collective count. Incoming
messages are handled as one ordered unit per channel, channels in
parallel (the C++ handlers release the GIL). Worker-side
barrier/allreduce ride a separate *gloo* group (host TCP), so they can
never interleave with the engine's NCCL traffic. Channels remain the
unit of key partitioning and per-channel protocol state (reference
--sys.channels), they just share the transport round.

On GPU the P2P exchange is dist.batch_isend_irecv = grouped
ncclSend/Recv over the xGMI point-to-point links (per-destination
batching, matching the reference's per-destination message builds,
sync_manager.h:305,361 — no rings: xGMI is per-link P2P). Staging
copies and the P2P ops run on a dedicated comm stream, event-ordered
against the default stream where the store kernels run, so transfers
overlap worker compute. On CPU it is gloo over loopback (the
multi-process no-GPU test tier).
"""
from __future__ import annotations

import math
import os
import sys
import threading
import time
from collections import defaultdict

import torch
import torch.distributed as dist

_TRACE_KEY = int(os.environ.get("ADAPM_TRACE_KEY", "-1"))

_BULK_CODES = {4: 2, 5: 1, 14: 2, 16: 2, 17: 3, 18: 2}  # code -> extra words per key


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
    sync_manager.h:62-105 estimate_sync_windows_and_tune): exponential
    smoothing (alpha=0.1) of the per-round clock advance, acted-ahead
    window = the 0.9999 quantile of Poisson(2*lambda) — exact inverse
    CDF for small lambda, normal approximation above 1e6 like the
    reference."""

    Q = 0.9999
    Z = 3.7190165  # Phi^-1(0.9999)

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

    @classmethod
    def poisson_quantile(cls, lam: float) -> int:
        if lam <= 0:
            return 0
        if lam > 500:  # normal approximation (exp(-lam) underflows past ~745;
            # at lam=500 the approximation is already within ~0.1%)
            return int(math.ceil(lam + cls.Z * math.sqrt(lam)))
        # exact inverse CDF walk
        p = math.exp(-lam)
        cdf = p
        k = 0
        while cdf < cls.Q and k < 10_000_000:
            k += 1
            p *= lam / k
            cdf += p
        return k

    def ahead(self) -> int:
        lam = self.rate * self.rounds_ahead
        # +small floor so cold-start intents are still acted on promptly
        return max(8, self.poisson_quantile(2.0 * lam))


class SyncManager:
    """Drives the single comm thread (all channels) of one rank."""

    def __init__(self, server, runtime, max_per_sec: float = 1000.0,
                 time_intent_actions: bool = True):
        self.server = server
        self.rt = runtime
        self.min_period = 1.0 / max_per_sec if max_per_sec > 0 else 0.0
        self.time_intent_actions = time_intent_actions
        self.stop_requested = threading.Event()
        self.kick_event = threading.Event()
        self.failed = False
        self.phase_totals = defaultdict(float)
        self.threads = []
        self.timer = ActionTimer()
        from concurrent.futures import ThreadPoolExecutor

        self._handler_pool = ThreadPoolExecutor(
            max_workers=max(1, runtime.num_channels - 1),
            thread_name_prefix="adapm-handler")
        self.watchdog_s = float(os.environ.get("ADAPM_WATCHDOG_S", "120"))
        self._progress = (0, time.monotonic())  # (total rounds, when it last moved)
        self._state = "init"  # coarse comm-thread position, for the watchdog dump
        # ADAPM_DEBUG_ROUNDS=N: every N rounds, dump outgoing record heads
        # (diagnoses livelocks where a record regenerates every round)
        self._dbg_every = int(os.environ.get("ADAPM_DEBUG_ROUNDS", "0"))
        if not time_intent_actions:
            server.set_intent_ahead(1 << 40)

    def start(self):
        if self.rt.world <= 1:
            return  # single rank: nothing to sync (reference sync_manager.h:454-457)
        t = threading.Thread(target=self._run, daemon=True, name="adapm-sync")
        t.start()
        self.threads.append(t)
        if self.watchdog_s > 0:
            w = threading.Thread(target=self._watchdog, daemon=True, name="adapm-watchdog")
            w.start()
            # not joined: daemon; it exits when stop is requested

    def request_stop(self):
        self.stop_requested.set()
        self.kick_event.set()

    def kick(self):
        """Wake the comm thread early: a worker enqueued remote ops and is
        (or will be) waiting on the round. All ranks under symmetric load
        kick at similar times, so the collective rounds speed up together;
        an early kicker just reaches the size all-gather sooner."""
        self.kick_event.set()

    def join(self):
        for t in self.threads:
            t.join()
        self.threads = []
        self._handler_pool.shutdown(wait=True)
        if os.environ.get("ADAPM_VERBOSE", "0") != "0" and self.phase_totals.get("rounds"):
            pt = dict(self.phase_totals)
            n = pt.pop("rounds")
            print(f"[adapm sync r{self.rt.rank}] {int(n)} rounds; per-round ms: " +
                  ", ".join(f"{k}={1000*v/n:.2f}" for k, v in sorted(pt.items())),
                  file=sys.stderr, flush=True)

    # ------------------------------------------------------------ watchdog

    def _watchdog(self):
        """Dump per-channel protocol state if rounds stop advancing while
        work is pending — a stalled collective should leave evidence, not
        a silently hung lease."""
        interval = min(10.0, self.watchdog_s / 4)
        while not self.stop_requested.wait(timeout=interval):
            if self.failed or not self.threads:
                return
            try:
                pend = self.server.debug_pending()
            except Exception:
                return
            total = sum(pend["rounds"])
            last_total, last_t = self._progress
            if total != last_total:
                self._progress = (total, time.monotonic())
                continue
            stalled_for = time.monotonic() - last_t
            if stalled_for > self.watchdog_s:
                busy = pend["tickets"] or any(pend["out_queues"]) or any(pend["responses"])
                print(f"[adapm WATCHDOG r{self.rt.rank}] sync rounds stalled "
                      f"{stalled_for:.0f}s ({'work pending' if busy else 'no queued work'}): "
                      f"rounds={pend['rounds']} tickets={pend['tickets']} "
                      f"out={pend['out_queues']} resp={pend['responses']} "
                      f"state={self._state}", file=sys.stderr, flush=True)
                self._progress = (total, time.monotonic())  # rate-limit the dump

    # ---------------------------------------------------------------- loop

    def _run(self):
        try:
            self._run_inner()
        except Exception as e:  # transport failure: a peer likely died
            print(f"[adapm] rank {self.rt.rank}: sync engine failed: "
                  f"{type(e).__name__}: {e}", file=sys.stderr, flush=True)
            self.failed = True
            self.server.fail(f"sync engine: {e}")

    def _run_inner(self):
        rt = self.rt
        nch = rt.num_channels
        # comm tensors live on the backend's device: GPU for NCCL (xGMI
        # P2P), CPU for gloo
        comm_dev = rt.device if rt.backend == "nccl" else torch.device("cpu")
        # ADAPM_FORCE_COMM_STREAM=1: exercise the comm-stream event
        # choreography on the gloo+CUDA test tier (otherwise it only
        # runs under NCCL, which needs one GPU per rank)
        use_streams = rt.is_cuda and (rt.backend == "nccl" or
                                      os.environ.get("ADAPM_FORCE_COMM_STREAM", "0") == "1")
        if rt.is_cuda:
            torch.cuda.set_device(rt.device)
        if use_streams:
            self._comm_stream = torch.cuda.Stream(rt.device)
            self._ev_fwd = torch.cuda.Event()
            self._ev_bwd = torch.cuda.Event()
        else:
            self._comm_stream = None

        verbose = os.environ.get("ADAPM_VERBOSE", "0") != "0"
        last_report = time.monotonic()
        rounds_at_report = 0
        n_rounds = 0
        while True:
            t0 = time.monotonic()
            if verbose and t0 - last_report > 10.0:
                clocks = self.server.worker_clocks()
                st = self.server.stats()
                rr, rp = st["replica_records"], st["replica_payloads"]
                pct = 100.0 * rp / rr if rr else 0.0
                print(f"[adapm sync r{rt.rank}] "
                      f"{(n_rounds - rounds_at_report) / (t0 - last_report):.0f} rounds/s, "
                      f"worker clocks {clocks}, {pct:.1f}% of replica records "
                      f"carried payload", flush=True)
                last_report, rounds_at_report = t0, n_rounds
            if self.time_intent_actions:
                self.server.set_intent_ahead(self.timer.update(self.server.worker_clocks()))

            stop = self.stop_requested.is_set()
            all_stopped, any_work = self._superround(nch, comm_dev, stop)
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

    def _superround(self, nch, comm_dev, stop_flag):
        s = self.server
        t0 = time.perf_counter()
        self._state = "collect"
        outs_a = [s.sync_collect(ch) for ch in range(nch)]
        t1 = time.perf_counter()
        self._state = "exchange_a"
        all_stopped, st_a = self._exchange(nch, comm_dev, outs_a, s.sync_process, stop_flag)
        t2 = time.perf_counter()
        self._state = "respond"
        outs_b = [s.sync_respond(ch) for ch in range(nch)]
        t3 = time.perf_counter()
        self._state = "exchange_b"
        _, st_b = self._exchange(nch, comm_dev, outs_b, s.sync_apply, stop_flag)
        t4 = time.perf_counter()
        self._state = "finish"
        any_work = False
        for ch in range(nch):
            a_meta, a_pay, a_local = st_a[ch]
            b_meta, b_pay, b_local = st_b[ch]
            # "Globally idle" for strong WaitSync = no DATA moved on this
            # channel anywhere: phase A carried no payload (payload-less
            # per-replica poll records are steady-state — every replica
            # announces its version each round, like the reference's
            # per-replica sync messages — and move no data) and phase B
            # carried nothing at all (any refresh/ack/response resets the
            # streak, so an in-flight request's answer keeps the round
            # busy). Every data-carrying record contributes payload on
            # every hop it travels, so an in-flight update always marks
            # its round non-idle.
            ch_idle = a_pay == 0 and a_local == 0 and b_meta == 0 and b_local == 0
            any_work = any_work or a_meta or a_pay or a_local or b_meta or b_local
            s.sync_finish(ch, ch_idle)
        pt = self.phase_totals
        pt["collect"] += t1 - t0
        pt["exchange_a"] += t2 - t1
        pt["respond"] += t3 - t2
        pt["exchange_b"] += t4 - t3
        pt["rounds"] += 1
        if self._dbg_every and int(pt["rounds"]) % self._dbg_every == 0:
            def heads(outs_all):
                return [[(dest, meta.reshape(-1)[:15].tolist()) for dest, meta, _ in outs]
                        for outs in outs_all]
            print(f"[adapm dbg r{self.rt.rank} round {int(pt['rounds'])}] "
                  f"stA={st_a} stB={st_b} outA={heads(outs_a)} outB={heads(outs_b)}",
                  file=sys.stderr, flush=True)
        return all_stopped, bool(any_work)

    def _exchange(self, nch, comm_dev, outs_per_ch, handler, stop_flag):
        """One phase for ALL channels: a single size all-gather (on its
        own process group) + one grouped P2P all-to-all-v where all
        channels' records to a peer are CONCATENATED into one meta and
        one payload message (so at most 2 same-pair messages are in
        flight per phase — gloo showed rare matching stalls with more,
        and interleaving collectives with P2P on one gloo context
        stalled too, hence the separate groups). Handlers then split by
        the per-channel sizes. Returns (all_stopped, per-channel (meta,
        payload, local) totals — the superround derives idleness)."""
        rt = self.rt
        world, rank = rt.world, rt.rank
        group = rt.sync_group
        use_streams = self._comm_stream is not None

        # size matrix row (per rank): nch x (n_meta, n_payload, local_work)
        # + one stop flag at the end
        sizes = torch.zeros(world * nch * 3 + 1, dtype=torch.int64)
        per_dest = [([], []) for _ in range(world)]  # (metas, payloads) in ch order
        local = []  # (ch, meta, payload) handled without transport
        for ch, outs in enumerate(outs_per_ch):
            for dest, meta, payload in outs:
                _trace(rank, ch, "out", dest, meta)
                if dest == rank:
                    local.append((ch, meta, payload))
                    sizes[(rank * nch + ch) * 3 + 2] = 1  # self-traffic: not idle
                    continue
                base = (dest * nch + ch) * 3
                sizes[base + 0] = meta.numel()
                sizes[base + 1] = payload.numel()
                per_dest[dest][0].append(meta.reshape(-1))
                per_dest[dest][1].append(payload)
        sizes[-1] = 1 if stop_flag else 0

        if use_streams:
            # comm stream must see the extract/gather kernels' writes to
            # the outgoing payloads (enqueued on the default stream)
            self._ev_fwd.record()
        stream_ctx = torch.cuda.stream(self._comm_stream) if use_streams else _nullctx()
        with stream_ctx:
            if use_streams:
                self._comm_stream.wait_event(self._ev_fwd)
            sizes_d = sizes.to(comm_dev, non_blocking=False)
            gathered = [torch.zeros_like(sizes_d) for _ in range(world)]
            self._state += ":allgather"
            dist.all_gather(gathered, sizes_d, group=rt.sizes_group)
            self._state += ":p2p"
            gathered = [g.cpu() for g in gathered]
            all_stopped = all(int(g[-1]) == 1 for g in gathered)

            # post sends/recvs in identical peer order on every rank;
            # exactly one meta (tag 0) + one payload (tag 1) per pair
            # per phase (tags are gloo-only; NCCL matches grouped P2P by
            # posting order and phases are fenced by the all-gather)
            use_tags = rt.backend != "nccl"
            p2p = []
            recv_bufs = {}
            for peer in range(world):
                if peer == rank:
                    continue
                n_meta = n_pay = 0
                for ch in range(nch):
                    base = (rank * nch + ch) * 3
                    n_meta += int(gathered[peer][base + 0])
                    n_pay += int(gathered[peer][base + 1])
                if n_meta > 0 or n_pay > 0:
                    rm = torch.empty(n_meta, dtype=torch.int64, device=comm_dev)
                    rp = torch.empty(n_pay, dtype=torch.float32, device=comm_dev)
                    recv_bufs[peer] = (rm, rp)
                    if n_meta:
                        p2p.append(dist.P2POp(dist.irecv, rm, peer, group,
                                              0 if use_tags else 0))
                    if n_pay:
                        p2p.append(dist.P2POp(dist.irecv, rp, peer, group,
                                              1 if use_tags else 0))
                metas, pays = per_dest[peer]
                if metas:
                    sm = (metas[0] if len(metas) == 1 else torch.cat(metas)).to(comm_dev)
                    sp = (pays[0] if len(pays) == 1 else torch.cat(pays))
                    if sp.device != comm_dev:
                        sp = sp.to(comm_dev)
                    if sm.numel():
                        p2p.append(dist.P2POp(dist.isend, sm, peer, group,
                                              0 if use_tags else 0))
                    if sp.numel():
                        p2p.append(dist.P2POp(dist.isend, sp, peer, group,
                                              1 if use_tags else 0))
            if p2p:
                reqs = dist.batch_isend_irecv(p2p)
                for r in reqs:
                    r.wait()
            if use_streams:
                self._ev_bwd.record(self._comm_stream)
        if use_streams:
            # store kernels (handlers run on the default stream) must see
            # the received payloads
            torch.cuda.default_stream(rt.device).wait_event(self._ev_bwd)

        # per-channel traffic totals over all (sender, dest) pairs
        gm = torch.stack([g[:-1] for g in gathered]).view(world, world, nch, 3)
        tot = gm.sum(dim=(0, 1))  # [nch, 3] = (meta, payload, local)
        stats = [(int(tot[ch, 0]), int(tot[ch, 1]), int(tot[ch, 2])) for ch in range(nch)]

        # split incoming messages per channel (local/self-targeted first,
        # then fixed peer order — the concatenation order on the sender)
        store_dev = rt.device
        th0 = time.perf_counter()
        per_ch = [[] for _ in range(nch)]
        for ch, meta, payload in local:
            per_ch[ch].append((rank, meta, payload))
        for peer in sorted(recv_bufs):
            rm, rp = recv_bufs[peer]
            meta_all = rm.cpu()
            payload_all = rp if rp.device == store_dev else rp.to(store_dev)
            mo = po = 0
            for ch in range(nch):
                base = (rank * nch + ch) * 3
                nm = int(gathered[peer][base + 0])
                npay = int(gathered[peer][base + 1])
                if nm == 0 and npay == 0:
                    continue
                meta = meta_all.narrow(0, mo, nm)
                payload = payload_all.narrow(0, po, npay)
                mo += nm
                po += npay
                _trace(rank, ch, "in ", peer, meta)
                per_ch[ch].append((peer, meta, payload))

        # run each channel's handlers as one ordered unit, channels in
        # parallel (channels partition the key space, and the C++
        # handlers release the GIL — this restores the per-channel
        # parallelism the round-1 thread-per-channel engine had, without
        # re-introducing its NCCL issuance hazards: all COMM still
        # happens on the single comm thread)
        def run_ch(ch):
            for peer, meta, payload in per_ch[ch]:
                handler(ch, peer, meta, payload)

        busy_chs = [ch for ch in range(nch) if per_ch[ch]]
        if len(busy_chs) > 1:
            futs = [self._handler_pool.submit(run_ch, ch) for ch in busy_chs[1:]]
            run_ch(busy_chs[0])
            for f in futs:
                f.result()
        elif busy_chs:
            run_ch(busy_chs[0])
        self.phase_totals["handlers"] += time.perf_counter() - th0
        return all_stopped, stats


class _nullctx:
    def __enter__(self):
        return None

    def __exit__(self, *a):
        return False
