"""Standalone symmetric-heap collectives: reduce_scatter and the
low-latency flag-in-payload allgather.

Capability parity (behavior only):
  * reduce_scatter — Triton-distributed
    python/triton_dist/kernels/nvidia/reduce_scatter.py:710-831.
  * ll_all_gather — the LL packed protocol of
    python/triton_dist/kernels/nvidia/low_latency_allgather.py:531-589
    (every 32-bit data word travels as an 8-byte (data, tag) pair; the
    receiver polls the payload itself — one xGMI crossing carries data
    AND signal, the right shape for small-message latency).

Both are hipGraph-capturable: the call tag is a monotonically bumped
device cell, so no flag resets are needed between calls.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap


@dataclass
class CollContext:
    heap: SymmHeap
    max_seg_elems: int       # reduce_scatter: output elems per rank
    max_ll_words: int        # ll_all_gather: 4-byte words per rank
    rs_inbox: SymmBuffer     # [world, max_seg_elems] bf16
    rs_flags: SymmBuffer     # [world * chunks] int32
    ll_inbox: SymmBuffer     # [world, max_ll_words, 2] int32
    tag_cell: torch.Tensor   # device int32 cell, bumped per call
    chunks: int

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_coll_context(max_seg_elems: int = 1 << 20,
                        max_ll_words: int = 1 << 14, chunks: int = 8,
                        heap: Optional[SymmHeap] = None) -> CollContext:
    heap = heap or get_heap()
    world = heap.world
    max_seg_elems = (max_seg_elems + 7) & ~7
    rs_inbox = heap.alloc_buffer((world, max_seg_elems), torch.bfloat16)
    rs_flags = heap.alloc_buffer((world * chunks,), torch.int32)
    ll_inbox = heap.alloc_buffer((world, max_ll_words, 2), torch.int32)
    # zero the protocol state once: tags start at 1 so stale garbage in a
    # fresh inbox can never match
    rs_flags.local().zero_()
    ll_inbox.local().zero_()
    if heap.backend == "hip":
        tag = torch.zeros(1, dtype=torch.int32, device="cuda")
        torch.cuda.synchronize()
        heap.barrier_all()
    else:
        tag = torch.zeros(1, dtype=torch.int32)
    return CollContext(heap, max_seg_elems, max_ll_words, rs_inbox,
                       rs_flags, ll_inbox, tag, chunks)


def reduce_scatter(x: torch.Tensor, ctx: CollContext,
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out[rank] = sum_r x_r[rank-th segment]. x: [world * m, ...] bf16,
    returns [m, ...]. m * row_elems must be a multiple of 8."""
    world, rank = ctx.world, ctx.rank
    assert x.shape[0] % world == 0
    seg_shape = (x.shape[0] // world,) + tuple(x.shape[1:])
    seg_elems = x.numel() // world
    if out is None:
        out = torch.empty(seg_shape, dtype=x.dtype, device=x.device)

    if ctx.heap.backend != "hip":
        # gloo has no reduce_scatter: all_reduce a copy, slice my segment
        full = x.float().contiguous()
        dist.all_reduce(full)
        out.copy_(full.reshape(world, -1)[rank].reshape(seg_shape)
                  .to(x.dtype))
        return out

    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    assert seg_elems % 8 == 0, "reduce_scatter: seg elems % 8 != 0"
    assert seg_elems <= ctx.max_seg_elems
    from .. import _C
    stream = torch.cuda.current_stream()
    _C.bump_cell(ctx.tag_cell.data_ptr(), stream.cuda_stream)
    _C.reduce_scatter(x.data_ptr(), ctx.rs_inbox.offset,
                      ctx.rs_flags.offset, ctx.rs_inbox.ptr(),
                      ctx.rs_flags.ptr(), out.data_ptr(), seg_elems,
                      ctx.chunks, ctx.tag_cell.data_ptr(),
                      stream.cuda_stream)
    return out


def ll_all_gather(x: torch.Tensor, ctx: CollContext,
                  out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Latency-optimized allgather for SMALL payloads (<= max_ll_words *
    4 bytes per rank). x: [m, ...]; returns [world * m, ...]."""
    world = ctx.world
    nbytes = x.numel() * x.element_size()
    assert nbytes % 4 == 0, "ll_all_gather: payload % 4 != 0"
    words = nbytes // 4
    out_shape = (world * x.shape[0],) + tuple(x.shape[1:])
    if out is None:
        out = torch.empty(out_shape, dtype=x.dtype, device=x.device)

    if ctx.heap.backend != "hip":
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x.contiguous())
        torch.cat(gathered, dim=0, out=out)
        return out

    assert x.is_contiguous()
    assert words <= ctx.max_ll_words, "payload too large for LL protocol"
    from .. import _C
    stream = torch.cuda.current_stream()
    _C.bump_cell(ctx.tag_cell.data_ptr(), stream.cuda_stream)
    _C.ll_allgather(x.data_ptr(), ctx.ll_inbox.offset, ctx.ll_inbox.ptr(),
                    out.data_ptr(), words, ctx.tag_cell.data_ptr(),
                    stream.cuda_stream)
    return out


def all_to_all_single(x: torch.Tensor, ctx: CollContext,
                      out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out[p-th segment] = rank p's x[my segment]: the classic
    all_to_all_single over the symmetric heap (push + flag-gated gather).
    x: [world * m, ...] bf16; segment elems must be a multiple of 8."""
    world = ctx.world
    assert x.shape[0] % world == 0
    seg_elems = x.numel() // world
    if out is None:
        out = torch.empty_like(x)

    if ctx.heap.backend != "hip":
        # gloo has no all_to_all: all_gather everything, pick my column
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x.contiguous())
        segs = [g.reshape(world, -1)[ctx.rank] for g in gathered]
        out.copy_(torch.stack(segs).reshape(x.shape))
        return out

    assert x.dtype == torch.bfloat16 and x.is_contiguous()
    assert seg_elems % 8 == 0 and seg_elems <= ctx.max_seg_elems
    from .. import _C
    stream = torch.cuda.current_stream()
    _C.bump_cell(ctx.tag_cell.data_ptr(), stream.cuda_stream)
    _C.all_to_all(x.data_ptr(), ctx.rs_inbox.offset, ctx.rs_flags.offset,
                  ctx.rs_inbox.ptr(), ctx.rs_flags.ptr(), out.data_ptr(),
                  seg_elems, ctx.chunks, ctx.tag_cell.data_ptr(),
                  stream.cuda_stream)
    return out


def reduce_scatter_ref(x: torch.Tensor, world: int,
                       rank: int) -> torch.Tensor:
    """Golden single-process reference (fp32 accumulate)."""
    segs = x.float().reshape(world, -1, *x.shape[1:])
    return segs.sum(0).to(x.dtype).reshape(x.shape[0] // world,
                                           *x.shape[1:])
