"""Local launcher: spawn N single-GPU (or CPU) ranks of a script on this
host — the rebuild of the reference tracker (reference
tracker/dmlc_local.py): env injection + keepalive restart on exit code
254 (reference dmlc_local.py:15-26). There is no scheduler process; the
torch.distributed TCPStore is the rendezvous.

    python -m adapm_amd.launch -n 4 [--keepalive] script.py [args...]
"""
from __future__ import annotations

import argparse
import os
import socket
import subprocess
import sys

KEEPALIVE_EXIT = 254


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", "--num-ranks", type=int, required=True)
    ap.add_argument("--master-addr", default="127.0.0.1")
    ap.add_argument("--master-port", type=int, default=0)
    ap.add_argument("--keepalive", action="store_true",
                    help="restart a rank that exits with code 254")
    ap.add_argument("cmd", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    if not args.cmd:
        ap.error("no command given")
    port = args.master_port or free_port()

    def spawn(rank):
        env = dict(os.environ)
        env.update(RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE=str(args.num_ranks),
                   MASTER_ADDR=args.master_addr, MASTER_PORT=str(port))
        env["PYTHONPATH"] = os.getcwd() + os.pathsep + env.get("PYTHONPATH", "")
        return subprocess.Popen([sys.executable] + args.cmd, env=env)

    procs = {r: spawn(r) for r in range(args.num_ranks)}
    code = 0
    try:
        pending = dict(procs)
        while pending:
            for r, p in list(pending.items()):
                rc = p.poll()
                if rc is None:
                    continue
                if rc == KEEPALIVE_EXIT and args.keepalive:
                    print(f"[launch] rank {r} exited 254; restarting", file=sys.stderr)
                    pending[r] = procs[r] = spawn(r)
                else:
                    del pending[r]
                    if rc != 0:
                        code = rc
                        for q in pending.values():
                            q.terminate()
                        pending.clear()
                        break
            import time

            time.sleep(0.2)
    finally:
        for p in procs.values():
            if p.poll() is None:
                p.terminate()
    sys.exit(code)


if __name__ == "__main__":
    main()
