"""Fused GEMM-ReduceScatter (intra-node, TP column-parallel epilogue).

MI355X-native redesign of the reference op (Triton-distributed
python/triton_dist/kernels/amd/gemm_reduce_scatter.py — capabilities:
producer GEMM whose epilogue stores each tile directly into the OWNER
rank's symmetric scatter buffer via remote xGMI stores (:128-227),
cross-GPU barrier, then a local ring-ordered reduce over the world
segments (:230-284)).

Here the producer is k_gemm_rs_producer_bf16 (csrc/kernels/gemm.hip): the
epilogue stages the C tile in LDS and issues 16B vectorized stores straight
over xGMI into owner_heap.scatter[src=my_rank] — scattered 2B remote
stores would waste the link. Reduce is k_rs_reduce_bf16.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap


@dataclass
class GemmRSContext:
    heap: SymmHeap
    max_m_per_rank: int
    n: int
    scatter: SymmBuffer  # [world, max_m_per_rank, N] bf16

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_gemm_rs_context(max_m_total: int, n: int,
                           heap: Optional[SymmHeap] = None) -> GemmRSContext:
    heap = heap or get_heap()
    world = heap.world
    assert max_m_total % world == 0
    m_per_rank = max_m_total // world
    scatter = heap.alloc_buffer((world, m_per_rank, n), torch.bfloat16)
    return GemmRSContext(heap, m_per_rank, n, scatter)


def gemm_rs(a: torch.Tensor, w: torch.Tensor, ctx: GemmRSContext,
            out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out[M/world, N] = ReduceScatter(A[M, K_shard] @ W[N, K_shard]^T)."""
    m, k = a.shape
    n = w.shape[0]
    world, rank = ctx.world, ctx.rank
    assert n == ctx.n and m % world == 0
    m_per_rank = m // world
    assert m_per_rank <= ctx.max_m_per_rank

    if ctx.heap.backend == "cpu":
        partial = (a.float() @ w.float().t())
        # scatter: copy my partial's segment for each owner into owner's buf
        ctx.heap.barrier_all()
        for owner in range(world):
            seg = partial[owner * m_per_rank:(owner + 1) * m_per_rank]
            ctx.scatter.peer(owner)[rank, :m_per_rank].copy_(
                seg.to(a.dtype))
        ctx.heap.barrier_all()
        acc = torch.zeros(m_per_rank, n, dtype=torch.float32)
        for s in range(world):
            r = (rank + 1 + s) % world
            acc += ctx.scatter.local()[r, :m_per_rank].float()
        res = acc.to(a.dtype)
        ctx.heap.barrier_all()  # reduce done before anyone's next scatter
        if out is not None:
            out.copy_(res)
            return out
        return res

    if world == 1:
        # no scatter/reduce at world 1: plain GEMM (hipBLASLt, or the
        # in-house split-K tier for K>>N shapes — see ops.gemm.best_gemm)
        from .gemm import best_gemm
        return best_gemm(a, w, out=out)

    heap, _C = ctx.heap, ctx.heap._C
    assert a.dtype == torch.bfloat16 and a.is_contiguous()
    assert m_per_rank % 128 == 0, "HIP path needs m/world % 128 == 0"
    compute = torch.cuda.current_stream()
    # entry barrier: previous call's reduce has consumed the scatter bufs
    heap.barrier_all_on_stream(compute)
    from .gemm import choose_splits, splitk_ws

    splits = choose_splits(m, n, k)
    if splits > 1:
        ws = splitk_ws(m, n, splits, a.device)
        _C.gemm_rs_producer_splitk_bf16(
            a.data_ptr(), w.data_ptr(), ws.data_ptr(), m, n, k,
            ctx.scatter.offset, m_per_rank, ctx.max_m_per_rank, world, rank,
            splits, compute.cuda_stream)
    else:
        _C.gemm_rs_producer_bf16(a.data_ptr(), w.data_ptr(), m, n, k,
                                 ctx.scatter.offset, m_per_rank,
                                 ctx.max_m_per_rank, world, rank,
                                 compute.cuda_stream)
    heap.barrier_all_on_stream(compute)
    if out is None:
        out = torch.empty(m_per_rank, n, dtype=torch.bfloat16,
                          device=a.device)
    _C.rs_reduce_bf16(ctx.scatter.ptr(), out.data_ptr(), world, rank,
                      m_per_rank, ctx.max_m_per_rank, n,
                      compute.cuda_stream)
    return out


def gemm_rs_ref(a: torch.Tensor, w: torch.Tensor, group=None) -> torch.Tensor:
    """Golden reference: matmul + torch.distributed reduce_scatter."""
    import torch.distributed as dist

    world = dist.get_world_size(group)
    partial = (a.float() @ w.float().t()).to(a.dtype).cpu().contiguous()
    out = torch.empty(a.shape[0] // world, w.shape[0], dtype=a.dtype)
    dist.reduce_scatter_tensor(out, partial, group=group)
    return out.to(a.device)
