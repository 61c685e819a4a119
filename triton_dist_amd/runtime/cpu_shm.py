"""CPU mock of the symmetric heap, backed by POSIX shared memory.

Gives the same one-sided put/get + notify/wait semantics as the hipIpc heap
so the distributed plumbing (BASELINE config 1: 2-rank tile notify/wait +
all-gather on gloo) is testable on CPU-only hosts. Aligned int32 stores on
x86 are atomic, so flag writes through numpy views behave like relaxed
atomics; `wait` spins with a short sleep.
"""
from __future__ import annotations

import os
import time
from multiprocessing import shared_memory

import numpy as np
import torch
import torch.distributed as dist

_TORCH_FROM_U8 = {
    torch.bfloat16: torch.bfloat16,
    torch.float16: torch.float16,
    torch.float32: torch.float32,
    torch.int32: torch.int32,
    torch.int64: torch.int64,
    torch.int8: torch.int8,
    torch.uint8: torch.uint8,
}


class CpuShmHeap:
    """Per-rank shared-memory block, all blocks mapped by every rank."""

    def __init__(self, group, size: int, uid: str | None = None):
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.size = size
        if uid is None:
            obj = [f"td{os.getpid()}_{time.time_ns() & 0xFFFFFF}"] \
                if self.rank == 0 else [None]
            if dist.is_initialized():
                dist.broadcast_object_list(obj, src=0, group=group)
            uid = obj[0]
        self.uid = uid
        self._own = shared_memory.SharedMemory(
            name=f"{uid}_r{self.rank}", create=True, size=size)
        self._own.buf[:] = b"\0" * size
        if dist.is_initialized():
            dist.barrier(group)
        self._blocks = []
        for r in range(self.world):
            if r == self.rank:
                self._blocks.append(self._own)
            else:
                self._blocks.append(
                    shared_memory.SharedMemory(name=f"{uid}_r{r}"))
        if dist.is_initialized():
            dist.barrier(group)

    def view(self, rank: int, offset: int, shape, dtype: torch.dtype
             ) -> torch.Tensor:
        nbytes = int(np.prod(shape)) * dtype.itemsize if shape else dtype.itemsize
        assert offset + nbytes <= self.size
        raw = np.frombuffer(self._blocks[rank].buf, dtype=np.uint8,
                            count=nbytes, offset=offset)
        t = torch.from_numpy(raw)
        return t.view(dtype).view(list(shape))

    def close(self):
        if dist.is_initialized():
            try:
                dist.barrier(self.group)
            except Exception:
                pass
        for r, b in enumerate(self._blocks):
            if r != self.rank:
                try:
                    b.close()
                except Exception:
                    pass
        try:
            self._own.close()
            self._own.unlink()
        except Exception:
            pass
        self._blocks = []


def notify(flag: torch.Tensor, index: int, val: int, add: bool = False):
    """Release-store (or add) an int32 flag in a peer's shm view."""
    assert flag.dtype == torch.int32
    if add:
        # not atomic cross-process: CPU mock restricts ADD signals to
        # single-writer-per-slot patterns (all our ops satisfy this).
        flag[index] = int(flag[index]) + val
    else:
        flag[index] = val


def wait_ge(flag: torch.Tensor, index: int, bound: int, timeout_s: float = 30.0):
    assert flag.dtype == torch.int32
    t0 = time.perf_counter()
    while int(flag[index]) < bound:
        time.sleep(1e-6)
        if time.perf_counter() - t0 > timeout_s:
            raise TimeoutError(
                f"wait_ge: flag[{index}]={int(flag[index])} < {bound} "
                f"after {timeout_s}s")


def wait_eq(flag: torch.Tensor, index: int, val: int, timeout_s: float = 30.0):
    assert flag.dtype == torch.int32
    t0 = time.perf_counter()
    while int(flag[index]) != val:
        time.sleep(1e-6)
        if time.perf_counter() - t0 > timeout_s:
            raise TimeoutError(
                f"wait_eq: flag[{index}]={int(flag[index])} != {val} "
                f"after {timeout_s}s")
