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


TILE = 256               # gemm_ar tile edge (gemm256 producer tile)
TILE_ELEMS = TILE * TILE


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
    # tile-granular fused GEMM+AR state (None when world == 1)
    tile_scatter: Optional[SymmBuffer] = None  # [world, slots, 256*256] bf16
    tile_out: Optional[SymmBuffer] = None      # [max_elems] bf16
    tile_arrive: Optional[SymmBuffer] = None   # [slots] int32
    tile_oflags: Optional[SymmBuffer] = None   # [tiles_max] int32
    tile_slots: int = 0
    comm_stream: Optional[object] = None
    ev_fork: Optional[object] = None
    ev_join: Optional[object] = None
    sk_ws: Optional[torch.Tensor] = None       # fp32 [max_elems] (lazy)
    sk_done: Optional[torch.Tensor] = None     # int32 [tiles_max] (lazy)

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
    ctx = AllReduceContext(heap, max_elems, chunks, inbox, outbox,
                           flags_in, flags_out)
    if world > 1:
        tiles_max = max(1, max_elems // TILE_ELEMS)
        slots = (tiles_max + world - 1) // world
        ctx.tile_scatter = heap.alloc_buffer((world, slots, TILE_ELEMS),
                                             torch.bfloat16)
        ctx.tile_out = heap.alloc_buffer((max_elems,), torch.bfloat16)
        ctx.tile_arrive = heap.alloc_buffer((slots,), torch.int32)
        ctx.tile_oflags = heap.alloc_buffer((tiles_max,), torch.int32)
        ctx.tile_slots = slots
        if heap.backend == "hip":
            ctx.comm_stream = torch.cuda.Stream()
            ctx.ev_fork = torch.cuda.Event()
            ctx.ev_join = torch.cuda.Event()
    return ctx


def ar_sk_pick(m: int, n: int, k: int) -> int:
    """Split-K factor for the gemm_ar 256^2 producer: fill the 256 CUs
    on occupancy-starved (decode) shapes while keeping >= 2 K-tiles per
    split so the ring pipeline has a steady state."""
    grid = (m // TILE) * (n // TILE)
    if grid >= 200:
        return 1
    best, best_d = 1, abs(grid - 256)
    for s in (2, 3, 4, 5, 6, 8, 9, 12, 16):
        if k % (128 * s) or (k // 128 // s) < 2:
            continue
        d = abs(grid * s - 256)
        if d < best_d:
            best, best_d = s, d
    return best


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
    """C = AllReduce(A @ W^T) — the gemm_ar TP mode's core op, tile-
    granular (capability parity with the reference's overlapped pair,
    kernels/amd/gemm_allreduce.py:104-200 — behavior only).

    Fused path (csrc/kernels/gemm_ar.hip): the producer GEMM pushes each
    256^2 C tile to its round-robin owner + arrive counter; a consumer on
    the comm stream reduces tiles as their `world` arrivals land and
    broadcasts into every rank's symmetric out buffer — AR cost hides
    under the GEMM tail. Falls back to GEMM-then-AR off the fast shapes.
    """
    m, k = a.shape
    n = w.shape[0]
    fuse = (ctx.world > 1 and ctx.tile_scatter is not None
            and m % TILE == 0 and n % TILE == 0 and k % 128 == 0
            and m * n <= ctx.max_elems)
    if ctx.heap.backend == "cpu":
        if fuse:
            return _gemm_ar_tiled_cpu(a, w, ctx, out)
        partial = (a.float() @ w.float().t()).to(a.dtype)
        return all_reduce(partial, ctx, out=out)
    if ctx.world == 1:
        from .gemm import best_gemm
        return best_gemm(a, w, out=out)
    if not fuse:
        partial = gemm(a, w)
        return all_reduce(partial, ctx, out=out, method=method)
    return _gemm_ar_tiled_hip(a, w, ctx, out)


def _n_owned(tiles: int, rank: int, world: int) -> int:
    return len(range(rank, tiles, world))


def _gemm_ar_tiled_hip(a: torch.Tensor, w: torch.Tensor,
                       ctx: AllReduceContext,
                       out: Optional[torch.Tensor],
                       sk: Optional[int] = None) -> torch.Tensor:
    assert a.dtype == torch.bfloat16 and a.is_contiguous()
    m, k = a.shape
    n = w.shape[0]
    world, rank = ctx.world, ctx.rank
    heap, _C = ctx.heap, ctx.heap._C
    compute = torch.cuda.current_stream()
    tiles = (m // TILE) * (n // TILE)
    s = compute.cuda_stream

    # reset my counters/flags; entry barrier proves every peer consumed
    # the previous call's tile_out and will only push after its barrier
    _C.reset_flags(ctx.tile_arrive.ptr(), ctx.tile_slots, 0, s)
    _C.reset_flags(ctx.tile_oflags.ptr(), tiles, 0, s)
    heap.barrier_all_on_stream(compute)

    ctx.ev_fork.record(compute)
    ctx.comm_stream.wait_event(ctx.ev_fork)

    if sk is None:
        sk = _gemm_ar_sk(a, w, ctx, m, n, k)
    ws_ptr = done_ptr = 0
    if sk > 1:
        if ctx.sk_ws is None or ctx.sk_ws.numel() < m * n:
            ctx.sk_ws = torch.empty(ctx.max_elems, dtype=torch.float32,
                                    device=a.device)
        if ctx.sk_done is None or ctx.sk_done.numel() < tiles:
            tiles_max = max(1, ctx.max_elems // TILE_ELEMS)
            ctx.sk_done = torch.empty(tiles_max, dtype=torch.int32,
                                      device=a.device)
        ws_ptr, done_ptr = ctx.sk_ws.data_ptr(), ctx.sk_done.data_ptr()

    _C.gemm_ar_producer_bf16(
        a.data_ptr(), w.data_ptr(), m, n, k, ctx.tile_scatter.offset,
        ctx.tile_arrive.offset, ctx.tile_out.offset, ctx.tile_oflags.offset,
        ctx.tile_slots, sk, ws_ptr, done_ptr, s)
    _C.ar_tile_consumer(
        m, n, ctx.tile_scatter.offset, ctx.tile_arrive.offset,
        ctx.tile_out.offset, ctx.tile_oflags.offset, ctx.tile_slots,
        _n_owned(tiles, rank, world), ctx.comm_stream.cuda_stream)
    ctx.ev_join.record(ctx.comm_stream)
    compute.wait_event(ctx.ev_join)
    # wait ALL tiles' flags — peers' consumers signal mine remotely
    _C.wait_eq(ctx.tile_oflags.ptr(), tiles, 1, s)
    res = ctx.tile_out.local()[:m * n].reshape(m, n)
    if out is not None:
        out.copy_(res)
        return out
    return res


_SK_TUNER = None


def _sk_tuner():
    global _SK_TUNER
    if _SK_TUNER is None:
        from ..tune import ContextualAutoTuner
        _SK_TUNER = ContextualAutoTuner(
            "gemm_ar_sk",
            [{"sk": s} for s in (1, 2, 3, 4, 5, 6, 8, 9, 12, 16)])
    return _SK_TUNER


def _gemm_ar_sk(a, w, ctx, m, n, k) -> int:
    """Producer split-K factor: tuned on hardware (invalid factors time
    out to inf inside the tuner); heuristic when capturing uncached."""
    key = f"m{m}_n{n}_k{k}_w{ctx.world}"
    t = _sk_tuner()._inner
    entry = t._mem.get(key)
    if entry:
        return entry["config"]["sk"]
    if torch.cuda.is_current_stream_capturing():
        return ar_sk_pick(m, n, k)

    def mk(c):
        sk = c["sk"]
        if sk > 1 and (k % (128 * sk) or (k // 128 // sk) < 2):
            raise ValueError("invalid sk for shape")
        return lambda: _gemm_ar_tiled_hip_sk(a, w, ctx, sk)

    return _sk_tuner().tune(key, mk)["sk"]


def _gemm_ar_tiled_hip_sk(a, w, ctx, sk):
    """Run the fused op with a FIXED sk (tuner closure)."""
    _gemm_ar_tiled_hip(a, w, ctx, None, sk=sk)


def _gemm_ar_tiled_cpu(a: torch.Tensor, w: torch.Tensor,
                       ctx: AllReduceContext,
                       out: Optional[torch.Tensor]) -> torch.Tensor:
    """CPU/gloo mirror of the tile protocol (same owner/slot math —
    barriers substitute the arrive counters)."""
    partial = (a.float() @ w.float().t()).to(a.dtype)
    m, n = partial.shape
    tn = n // TILE
    tiles = (m // TILE) * tn
    world, rank = ctx.world, ctx.rank
    ctx.heap.barrier_all()
    for lt in range(tiles):
        o, slot = lt % world, lt // world
        pm, pn = divmod(lt, tn)
        tile = partial[pm * TILE:(pm + 1) * TILE,
                       pn * TILE:(pn + 1) * TILE].reshape(-1)
        ctx.tile_scatter.peer(o)[rank, slot].copy_(tile)
    ctx.heap.barrier_all()
    for slot, lt in enumerate(range(rank, tiles, world)):
        pm, pn = divmod(lt, tn)
        acc = ctx.tile_scatter.local()[:, slot].float().sum(0)
        red = acc.to(a.dtype).reshape(TILE, TILE)
        for p in range(world):
            ov = ctx.tile_out.peer(p)[:m * n].reshape(m, n)
            ov[pm * TILE:(pm + 1) * TILE,
               pn * TILE:(pn + 1) * TILE].copy_(red)
    ctx.heap.barrier_all()
    res = ctx.tile_out.local()[:m * n].reshape(m, n).clone()
    if out is not None:
        out.copy_(res)
        return out
    return res


def all_reduce_ref(x: torch.Tensor, group=None) -> torch.Tensor:
    import torch.distributed as dist

    cpu = x.detach().float().cpu()
    dist.all_reduce(cpu, group=group)
    return cpu.to(x.dtype).to(x.device)
