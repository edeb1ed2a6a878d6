"""Process/rank runtime: torch.distributed init + process groups.

Replaces the reference's Postoffice/Van node management (reference
src/postoffice.cc, src/van.cc): rendezvous comes from torchrun-style env
vars (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT) with DMLC_* fallbacks; the
scheduler process disappears (TCPStore rendezvous replaces ADD_NODE).

One process per GPU. Channels map to dedicated process groups so each
channel's sync thread can run collectives concurrently (RCCL over xGMI on
GPU, gloo on CPU).
"""
from __future__ import annotations

import datetime
import os
import threading
from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist


def _env(name, *alts, default=None):
    for n in (name,) + alts:
        v = os.environ.get(n)
        if v is not None:
            return v
    return default


@dataclass
class Runtime:
    rank: int
    world: int
    device: torch.device
    backend: str
    num_channels: int
    # The sync engine is driven by ONE comm thread per rank, issuing on
    # TWO process groups in a fixed alternation (size all-gather on
    # sizes_group, batched P2P on sync_group). With NCCL, the single
    # issuing thread makes the cross-communicator order deterministic on
    # every rank; with gloo, separating the collective from the P2P
    # traffic keeps them on different contexts (interleaving both on one
    # gloo context produced rare matching stalls).
    sync_group: Optional[object] = None
    sizes_group: Optional[object] = None
    # Worker-side barrier/allreduce ride a SEPARATE gloo group (host TCP):
    # tiny control-plane collectives that must never interleave with the
    # sync engine's NCCL traffic (they are called from worker threads,
    # concurrently with sync rounds).
    worker_group: Optional[object] = None
    worker_group_lock: threading.Lock = field(default_factory=threading.Lock)

    @property
    def is_cuda(self) -> bool:
        return self.device.type == "cuda"


_RUNTIME: Optional[Runtime] = None


def init_runtime(num_channels: int = 2, device: str | None = None,
                 timeout_s: float = 180.0) -> Runtime:
    """Initialize torch.distributed (if world>1) and per-channel groups.

    device: "cpu", "cuda", or None (auto: cuda:{LOCAL_RANK} if available).
    """
    global _RUNTIME
    if _RUNTIME is not None:
        return _RUNTIME

    rank = int(_env("RANK", "DMLC_RANK", default="0"))
    world = int(_env("WORLD_SIZE", "DMLC_NUM_SERVER", default="1"))
    local_rank = int(_env("LOCAL_RANK", default=str(rank)))

    if device is None:
        device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
    dev = torch.device(device)
    if dev.type == "cuda":
        torch.cuda.set_device(dev)

    backend = "nccl" if dev.type == "cuda" else "gloo"
    if os.environ.get("ADAPM_FORCE_GLOO", "0") == "1":
        backend = "gloo"  # e.g. several ranks sharing one GPU for testing
    if world > 1:
        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", _env("DMLC_PS_ROOT_URI", default="127.0.0.1"))
            os.environ.setdefault("MASTER_PORT", _env("DMLC_PS_ROOT_PORT", default="29500"))
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        sync_group = dist.new_group(backend=backend)
        sizes_group = dist.new_group(backend=backend)
        worker_group = dist.new_group(backend="gloo")
    else:
        sync_group = None
        sizes_group = None
        worker_group = None

    _RUNTIME = Runtime(rank=rank, world=world, device=dev, backend=backend,
                       num_channels=num_channels, sync_group=sync_group,
                       sizes_group=sizes_group, worker_group=worker_group)
    return _RUNTIME


def get_runtime() -> Runtime:
    if _RUNTIME is None:
        raise RuntimeError("adapm_amd runtime not initialized — call adapm_amd.setup() first")
    return _RUNTIME


def shutdown_runtime():
    global _RUNTIME
    if _RUNTIME is not None and _RUNTIME.world > 1 and dist.is_initialized():
        dist.destroy_process_group()
    _RUNTIME = None
