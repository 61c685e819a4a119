"""Point-to-point ops over the symmetric heap (pipeline-parallel comm).

Capability parity with the reference's P2P/PP layer (Triton-distributed
kernels/nvidia/p2p.py:33-119 p2p_set_signal/wait/put/copy and
layers/nvidia/pp_block.py:36-149 PPCommLayer — behavior only): SPSC ring
slots per (src, dst) pair with credit-based flow control, SDMA data + 4B
flag signals.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch

from ..runtime import cpu_shm
from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap


@dataclass
class P2PContext:
    heap: SymmHeap
    max_bytes: int
    depth: int
    inbox: SymmBuffer   # [world, depth, max_bytes] uint8 (indexed by src)
    full: SymmBuffer    # [world, depth] int32 — msg seq number when full
    credit: SymmBuffer  # [world, depth] int32 — consumer acks (indexed by dst)
    sent: dict = field(default_factory=dict)   # per-dst send seq
    rcvd: dict = field(default_factory=dict)   # per-src recv seq

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_p2p_context(max_bytes: int, depth: int = 2,
                       heap: Optional[SymmHeap] = None) -> P2PContext:
    heap = heap or get_heap()
    w = heap.world
    return P2PContext(
        heap, max_bytes, depth,
        inbox=heap.alloc_buffer((w, depth, max_bytes), torch.uint8),
        full=heap.alloc_buffer((w, depth), torch.int32),
        credit=heap.alloc_buffer((w, depth), torch.int32),
    )


def p2p_send(x: torch.Tensor, dst: int, ctx: P2PContext):
    """Stream-ordered send of a contiguous tensor to rank `dst`."""
    nbytes = x.numel() * x.element_size()
    assert nbytes <= ctx.max_bytes and x.is_contiguous()
    seq = ctx.sent.get(dst, 0)
    slot = seq % ctx.depth
    heap, rank = ctx.heap, ctx.rank
    if heap.backend == "cpu":
        if seq >= ctx.depth:  # wait for consumer credit on this slot
            cpu_shm.wait_ge(ctx.credit.local()[dst], slot,
                            seq // ctx.depth)
        ctx.inbox.peer(dst)[rank, slot, :nbytes].copy_(
            x.reshape(-1).view(torch.uint8))
        cpu_shm.notify(ctx.full.peer(dst)[rank], slot, seq + 1)
    else:
        _C = heap._C
        s = torch.cuda.current_stream().cuda_stream
        if seq >= ctx.depth:
            _C.wait_eq(ctx.credit.ptr() + (dst * ctx.depth + slot) * 4, 1,
                       seq // ctx.depth, s)
        dst_ptr = ctx.inbox.ptr(dst) + \
            (rank * ctx.depth + slot) * ctx.max_bytes
        _C.memcpy_async(dst_ptr, x.data_ptr(), nbytes, s)
        _C.signal_set(ctx.full.ptr(dst) + (rank * ctx.depth + slot) * 4,
                      seq + 1, s)
    ctx.sent[dst] = seq + 1


def p2p_recv(out: torch.Tensor, src: int, ctx: P2PContext) -> torch.Tensor:
    """Stream-ordered receive of a contiguous tensor from rank `src`."""
    nbytes = out.numel() * out.element_size()
    assert nbytes <= ctx.max_bytes and out.is_contiguous()
    seq = ctx.rcvd.get(src, 0)
    slot = seq % ctx.depth
    heap = ctx.heap
    if heap.backend == "cpu":
        cpu_shm.wait_ge(ctx.full.local()[src], slot, seq + 1)
        out.reshape(-1).view(torch.uint8).copy_(
            ctx.inbox.local()[src, slot, :nbytes])
        cpu_shm.notify(ctx.credit.peer(src)[ctx.rank], slot,
                       seq // ctx.depth + 1)
    else:
        _C = heap._C
        s = torch.cuda.current_stream().cuda_stream
        _C.wait_eq(ctx.full.ptr() + (src * ctx.depth + slot) * 4, 1,
                   seq + 1, s)
        src_ptr = ctx.inbox.ptr() + (src * ctx.depth + slot) * ctx.max_bytes
        _C.memcpy_async(out.data_ptr(), src_ptr, nbytes, s)
        # credit back after the copy (stream-ordered)
        _C.signal_set(ctx.credit.ptr(src) + (ctx.rank * ctx.depth + slot) * 4,
                      seq // ctx.depth + 1, s)
    ctx.rcvd[src] = seq + 1
    return out


class PPCommLayer:
    """Pipeline-stage communication: send activations forward / receive from
    the previous stage (reference layers/nvidia/pp_block.py capability)."""

    def __init__(self, ctx: P2PContext, stage: int, n_stages: int):
        self.ctx = ctx
        self.stage = stage
        self.n_stages = n_stages

    @property
    def prev(self):
        return (self.stage - 1) % self.n_stages

    @property
    def next(self):
        return (self.stage + 1) % self.n_stages

    def send_forward(self, x: torch.Tensor):
        if self.stage < self.n_stages - 1:
            p2p_send(x, self.next, self.ctx)

    def recv_forward(self, out: torch.Tensor):
        if self.stage > 0:
            p2p_recv(out, self.prev, self.ctx)
        return out
