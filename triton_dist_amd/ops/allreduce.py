"""AllReduce ops over the symmetric heap + the fused GEMM-AllReduce op.

Capability parity with the reference AR family (Triton-distributed
python/triton_dist/kernels/allreduce.py:31-49 AllReduceMethod,
kernels/amd/gemm_allreduce.py gemm_allreduce_op — behavior only).
Methods: one_shot (latency, full-mesh push over the 7 xGMI links),
two_shot (bandwidth, reduce-scatter + broadcast), auto by size.
Includes the reference's straggler-injection hook for jitter testing.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from ..runtime import cpu_shm
from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap
from .gemm import gemm

TWO_SHOT_THRESHOLD = 4 << 20  # bytes; above this two_shot wins on xGMI


@dataclass
class AllReduceContext:
    heap: SymmHeap
    max_elems: int
    chunks: int
    inbox: SymmBuffer     # [world, max_elems] bf16
    outbox: SymmBuffer    # [world, max_elems/world…] reuse of inbox-sized buf
    flags_in: SymmBuffer  # [world * chunks] int32
    flags_out: SymmBuffer
    epoch: int = 0

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_allreduce_context(max_elems: int, chunks: int = 0,
                             heap: Optional[SymmHeap] = None
                             ) -> AllReduceContext:
    heap = heap or get_heap()
    world = heap.world
    if chunks <= 0:
        chunks = max(8, min(1024, (max_elems * 2) // 65536))
    inbox = heap.alloc_buffer((world, max_elems), torch.bfloat16)
    outbox = heap.alloc_buffer((world, (max_elems + world - 1) // world),
                               torch.bfloat16)
    flags_in = heap.alloc_buffer((world * chunks,), torch.int32)
    flags_out = heap.alloc_buffer((world * chunks,), torch.int32)
    return AllReduceContext(heap, max_elems, chunks, inbox, outbox,
                            flags_in, flags_out)


def all_reduce(x: torch.Tensor, ctx: AllReduceContext,
               out: Optional[torch.Tensor] = None, method: str = "auto",
               straggler_rank: int = -1, straggler_cycles: int = 0
               ) -> torch.Tensor:
    """Sum-allreduce of a bf16 tensor across the node."""
    elems = x.numel()
    assert elems <= ctx.max_elems
    world, rank = ctx.world, ctx.rank
    if out is None:
        out = torch.empty_like(x)

    if ctx.heap.backend == "cpu":
        ctx.epoch += 1
        ctx.heap.barrier_all()
        for p in range(world):
            ctx.inbox.peer(p)[rank, :elems].copy_(x.reshape(-1))
            for c in range(ctx.chunks):
                cpu_shm.notify(ctx.flags_in.peer(p), rank * ctx.chunks + c,
                               ctx.epoch)
        fl = ctx.flags_in.local()
        for i in range(world * ctx.chunks):
            cpu_shm.wait_ge(fl, i, ctx.epoch)
        acc = ctx.inbox.local()[:, :elems].float().sum(0)
        out.copy_(acc.to(x.dtype).reshape(x.shape))
        return out

    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    heap, _C = ctx.heap, ctx.heap._C
    stream = torch.cuda.current_stream()
    if method == "auto":
        method = "two_shot" if (elems * 2 > TWO_SHOT_THRESHOLD
                                and elems % (8 * world) == 0
                                and world > 1) else "one_shot"
    # reset + entry barrier (graph-safe constants, same pattern as AG)
    _C.reset_flags(ctx.flags_in.ptr(), world * ctx.chunks, 0,
                   stream.cuda_stream)
    _C.reset_flags(ctx.flags_out.ptr(), world * ctx.chunks, 0,
                   stream.cuda_stream)
    heap.barrier_all_on_stream(stream)
    if method == "one_shot":
        _C.allreduce_oneshot(x.data_ptr(), out.data_ptr(),
                             ctx.inbox.offset, ctx.flags_in.offset, elems,
                             ctx.chunks, straggler_rank, straggler_cycles,
                             stream.cuda_stream)
    else:
        _C.allreduce_twoshot(x.data_ptr(), out.data_ptr(),
                             ctx.inbox.offset, ctx.outbox.offset,
                             ctx.flags_in.offset, ctx.flags_out.offset,
                             elems, ctx.chunks, stream.cuda_stream)
    return out


def gemm_allreduce(a: torch.Tensor, w: torch.Tensor,
                   ctx: AllReduceContext,
                   out: Optional[torch.Tensor] = None,
                   method: str = "auto") -> torch.Tensor:
    """C = AllReduce(A @ W^T) — the gemm_ar TP mode's core op (reference
    kernels/amd/gemm_allreduce.py:104-200 capability; v1 runs the persistent
    GEMM then the AR kernels on-stream — tile-granular notify overlap is a
    planned refinement)."""
    m, n = a.shape[0], w.shape[0]
    if ctx.heap.backend == "cpu":
        partial = (a.float() @ w.float().t()).to(a.dtype)
        return all_reduce(partial, ctx, out=out)
    partial = gemm(a, w)
    return all_reduce(partial, ctx, out=out, method=method)


def all_reduce_ref(x: torch.Tensor, group=None) -> torch.Tensor:
    import torch.distributed as dist

    cpu = x.detach().float().cpu()
    dist.all_reduce(cpu, group=group)
    return cpu.to(x.dtype).to(x.device)
