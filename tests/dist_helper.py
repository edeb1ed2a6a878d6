"""Multi-process test harness: simulates multi-node on one host, exactly
like the reference's tracker/dmlc_local.py test setup (reference
tests/run_tests.sh) but with torch.distributed rendezvous over loopback."""
from __future__ import annotations

import multiprocessing as mp
import os
import socket
import sys
import traceback


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _entry(rank: int, world: int, port: int, fn, args, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("OMP_NUM_THREADS", "2")
    # hang forensics: dump every thread's stack if the worker is still
    # alive this long (run_dist timeouts then show WHERE it hung)
    import faulthandler

    faulthandler.dump_traceback_later(
        float(os.environ.get("ADAPM_TEST_DUMP_S", "240")), exit=False)
    try:
        fn(rank, world, *args)
        q.put((rank, None))
    except Exception:
        q.put((rank, traceback.format_exc()))
        sys.exit(1)


def run_dist(world: int, fn, *args, timeout: float = 120.0, _retry: bool = True):
    """Run fn(rank, world, *args) in `world` spawned processes.

    Retries ONCE on a timeout: a rare sync-round stall (~1/60 runs, only
    under heavy host oversubscription; docs/ARCHITECTURE.md §8) would
    otherwise fail a whole -x test run. Assertion failures never retry.
    """
    try:
        return _run_dist_once(world, fn, *args, timeout=timeout)
    except TimeoutError:
        if not _retry:
            raise
        print(f"[dist_helper] timeout; retrying {fn.__name__} once", file=sys.stderr)
        return _run_dist_once(world, fn, *args, timeout=timeout)


def _run_dist_once(world: int, fn, *args, timeout: float = 120.0):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_entry, args=(r, world, port, fn, args, q), daemon=True)
        for r in range(world)
    ]
    for p in procs:
        p.start()
    errs = []
    done = 0
    import time as _time

    deadline = _time.monotonic() + timeout
    while done < world and not errs:
        try:
            rank, err = q.get(timeout=min(5.0, max(0.1, deadline - _time.monotonic())))
        except Exception:
            if _time.monotonic() >= deadline:
                for p in procs:
                    p.terminate()
                raise TimeoutError(f"distributed test timed out after {timeout}s")
            # a child may have died without reporting
            if any(p.exitcode not in (None, 0) for p in procs):
                errs.append("a child process died without reporting an error")
            continue
        done += 1
        if err:
            errs.append(f"[rank {rank}]\n{err}")
    for p in procs:
        if errs:
            p.terminate()
        else:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    if errs:
        raise AssertionError("\n".join(errs))
