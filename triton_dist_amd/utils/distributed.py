"""Process bootstrap and distributed helpers.

MI355X-native counterpart of the reference runtime layer
(Triton-distributed python/triton_dist/utils.py:341-372
`initialize_distributed`, :302 `finalize_distributed`, :445 `dist_print`):
one process per GPU, torch.distributed with cpu:gloo + cuda:nccl (RCCL on
ROCm) for bootstrap and golden-reference collectives; the data plane is the
hipIpc symmetric heap (runtime/symm_mem.py), not RCCL.
"""
from __future__ import annotations

import datetime
import os
import random
import sys
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

_INITIALIZED = False


def has_gpu() -> bool:
    return torch.cuda.is_available()


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", str(env_rank())))


def local_device() -> int:
    """Device index for this rank. Clamped modulo visible devices so a
    2-process world can share one GPU (IPC paths are testable on a 1-GPU
    box; xGMI paths then light up unchanged on 8)."""
    n = torch.cuda.device_count()
    return env_local_rank() % max(n, 1)


_SHARED_GPU: Optional[bool] = None


def gpu_oversubscribed(world: Optional[int] = None) -> bool:
    """True when two ranks share one PHYSICAL GPU (a validation box
    running a 2-rank world on one device). Two processes' large
    spin-wait grids can then occupy every CU slot and starve each
    other's producer/dispatch kernels until the 30 s spin watchdog
    traps — callers must route to non-spinning (or single-workgroup-
    spin) fallbacks. On a one-rank-per-GPU deployment this is always
    False and the overlap paths run.

    `world > device_count` alone is ambiguous: a launcher that pins one
    visible device per rank (HIP_VISIBLE_DEVICES) also shows count 1 at
    world 8 without any sharing. When the group is up, ranks exchange
    their device's (host, PCI domain/bus/device) once over gloo and
    look for duplicates; the env heuristic is only the fallback."""
    if not has_gpu():
        return False
    w = world if world is not None else env_world_size()
    if w <= torch.cuda.device_count():
        return False
    if not dist.is_initialized():
        return True  # conservative: cannot verify isolation
    global _SHARED_GPU
    if _SHARED_GPU is None:
        import socket

        p = torch.cuda.get_device_properties(torch.cuda.current_device())
        mine = (socket.gethostname(), getattr(p, "pci_domain_id", -1),
                getattr(p, "pci_bus_id", -1),
                getattr(p, "pci_device_id", -1))
        allv = [None] * dist.get_world_size()
        dist.all_gather_object(allv, mine)
        _SHARED_GPU = len(set(allv)) < len(allv)
    return _SHARED_GPU


def initialize_distributed(seed: int = 42, timeout_s: int = 1800,
                           backend: Optional[str] = None):
    """Init the process group (gloo on CPU-only hosts, gloo+RCCL on GPU),
    pin the device for this rank, and seed all RNGs identically per rank.

    Returns the default process group.
    """
    global _INITIALIZED
    if _INITIALIZED:
        return dist.group.WORLD
    rank, world = env_rank(), env_world_size()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend is None:
        # RCCL needs one rank per device; when ranks share a GPU (1-GPU box
        # running a 2-rank IPC test) bootstrap over gloo only.
        if has_gpu() and torch.cuda.device_count() >= world:
            backend = "cpu:gloo,cuda:nccl"
        else:
            backend = "gloo"
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    if has_gpu():
        torch.cuda.set_device(local_device())
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed + rank)
    if has_gpu():
        torch.cuda.manual_seed_all(seed + rank)
    _INITIALIZED = True
    return dist.group.WORLD


def finalize_distributed():
    global _INITIALIZED, _SHARED_GPU
    _SHARED_GPU = None
    from ..runtime.symm_mem import shutdown_heap

    shutdown_heap()
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()
    _INITIALIZED = False


def rank(group=None) -> int:
    return dist.get_rank(group) if dist.is_initialized() else 0


def world_size(group=None) -> int:
    return dist.get_world_size(group) if dist.is_initialized() else 1


def dist_print(*args, allowed_ranks="0", need_sync: bool = False, **kwargs):
    """Rank-filtered printing (cf. utils.py:445-476 semantics).

    allowed_ranks: "all" or iterable/str of ranks.
    """
    r, w = rank(), world_size()
    if allowed_ranks == "all":
        allowed = list(range(w))
    elif isinstance(allowed_ranks, str):
        allowed = [int(x) for x in allowed_ranks.split(",") if x != ""]
    else:
        allowed = list(allowed_ranks)
    for i in range(w):
        if need_sync and dist.is_initialized():
            dist.barrier()
        if i == r and r in allowed:
            print(f"[rank {r}]", *args, **kwargs)
            sys.stdout.flush()
