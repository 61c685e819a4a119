"""TP-MoE: AllGather + grouped GEMM and grouped GEMM + topk-reduce +
ReduceScatter — the tensor-parallel MoE path (experts replicated, each
rank holding an intermediate-dim shard of EVERY expert).

Capability parity (behavior only) with Triton-distributed's MoE-TP pair:
  * AG + GroupGEMM  — kernels/nvidia/allgather_group_gemm.py /
    group_gemm.py, plus the moe_ag_scatter_align sort
    (csrc/lib/moe_utils.cu:61-356).
  * GroupGEMM + RS  — kernels/nvidia/moe_reduce_rs.py:168-730
    (grouped GEMM -> weighted topk reduce -> reduce_scatter).

MI355X design: the gather rides the symmetric-heap push allgather, the
expert-sorted grouped GEMMs ride the same persistent work-queue kernel
as the EP path (csrc/kernels/moe.hip k_moe_grouped_gemm_pq), the final
cross-rank sum of intermediate-shard partials rides the standalone
reduce_scatter (flag-in-payload free: chunked release flags). The
token sort itself is index arithmetic (argsort + index_select), not a
custom kernel — CDNA4 has no cheap global atomics win over torch's sort
at prefill sizes, and this op is not on the decode hot path.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist

from ..runtime.symm_mem import get_heap


def moe_sort_tokens(x_full: torch.Tensor, topk_ids: torch.Tensor,
                    n_experts: int, bm: int = 32
                    ) -> Tuple[torch.Tensor, torch.Tensor, dict]:
    """Expert-sort the gathered tokens: returns (x_sorted [M*K+slack, H],
    tok_index [M*K] (source token of each sorted row), meta dict with the
    grouped-GEMM descriptors expert_base/expert_rows/work_items/
    work_count, all int32 on device)."""
    mk = topk_ids.numel()
    k = topk_ids.shape[1]
    flat = topk_ids.reshape(-1).to(torch.int64)
    order = torch.argsort(flat, stable=True)
    tok = order // k
    x_sorted = torch.zeros(mk + 128, x_full.shape[1], dtype=x_full.dtype,
                           device=x_full.device)
    x_sorted[:mk] = x_full.index_select(0, tok)
    counts = torch.bincount(flat, minlength=n_experts)
    base = torch.cumsum(counts, 0) - counts
    cnt = counts.cpu().tolist()
    items = []
    for e, c in enumerate(cnt):
        for t in range((c + bm - 1) // bm):
            items.append(e * 65536 + t)
    dev = x_full.device
    meta = {
        "expert_base": base.to(torch.int32).to(dev).contiguous(),
        "expert_rows": counts.to(torch.int32).to(dev).contiguous(),
        "work_items": torch.tensor(items or [0], dtype=torch.int32,
                                   device=dev),
        "work_count": torch.tensor([len(items)], dtype=torch.int32,
                                   device=dev),
        "order": order,
    }
    return x_sorted, tok, meta


def grouped_gemm(x_sorted: torch.Tensor, weights: torch.Tensor,
                 meta: dict, out: Optional[torch.Tensor] = None
                 ) -> torch.Tensor:
    """out[rows of expert e] = x_sorted[rows] @ weights[e]^T via the
    persistent work-queue kernel. weights: [E, N, K]; N % 128 == 0,
    K % 64 == 0."""
    n = weights.shape[1]
    kk = weights.shape[2]
    if out is None:
        out = torch.empty(x_sorted.shape[0], n, dtype=x_sorted.dtype,
                          device=x_sorted.device)
    if not x_sorted.is_cuda:
        eb = meta["expert_base"].tolist()
        er = meta["expert_rows"].tolist()
        for e in range(weights.shape[0]):
            lo, r = eb[e], er[e]
            if r:
                out[lo:lo + r] = (x_sorted[lo:lo + r].float()
                                  @ weights[e].float().t()).to(out.dtype)
        return out
    from .. import _C
    s = torch.cuda.current_stream().cuda_stream
    _C.moe_grouped_gemm_pq(x_sorted.data_ptr(), weights.data_ptr(),
                           out.data_ptr(), meta["expert_base"].data_ptr(),
                           meta["expert_rows"].data_ptr(),
                           meta["work_items"].data_ptr(),
                           meta["work_count"].data_ptr(), n, kk, s)
    return out


def tp_moe_from_full(x_full: torch.Tensor, topk_ids: torch.Tensor,
                     topk_w: torch.Tensor, w_gate_up: torch.Tensor,
                     w_down: torch.Tensor, coll_ctx) -> torch.Tensor:
    """The compute half of TP-MoE, given the already-gathered batch:
    sorted grouped gate/up -> SwiGLU -> grouped down (intermediate-shard
    partials) -> weighted topk reduce -> reduce_scatter."""
    from .collectives import reduce_scatter
    from .fused import swiglu_op

    heap = get_heap()
    world = heap.world
    e_num = w_gate_up.shape[0]
    inter_shard = w_gate_up.shape[1] // 2
    h = x_full.shape[1]
    mk = topk_ids.numel()
    x_sorted, tok, meta = moe_sort_tokens(x_full, topk_ids, e_num)

    hidden = grouped_gemm(x_sorted, w_gate_up, meta)
    act = swiglu_op(hidden, inter_shard)
    part = grouped_gemm(act, w_down, meta)

    w_sorted = topk_w.reshape(-1)[meta["order"]].to(torch.float32)
    y_full = torch.zeros(x_full.shape[0], h, dtype=torch.float32,
                         device=x_full.device)
    y_full.index_add_(0, tok, part[:mk].float() * w_sorted[:, None])
    y_full = y_full.to(x_full.dtype)

    if world == 1:
        return y_full
    return reduce_scatter(y_full, coll_ctx)


def tp_moe_forward(x_shard: torch.Tensor, topk_ids: torch.Tensor,
                   topk_w: torch.Tensor, w_gate_up: torch.Tensor,
                   w_down: torch.Tensor, ag_ctx, coll_ctx,
                   overlap: Optional[bool] = None) -> torch.Tensor:
    """Full TP-MoE block: AG(x_shard) -> tp_moe_from_full. topk_ids /
    topk_w are for the FULL gathered batch [M, K] (router is replicated +
    deterministic).

    w_gate_up: [E, 2*inter_shard, H] (my shard of gate rows then up rows),
    w_down: [E, H, inter_shard]. Returns [m_local, H].

    overlap (default on for hip, world > 1): segment-progressive
    AG↔grouped-GEMM overlap — see _tp_moe_forward_overlapped."""
    from .allgather_gemm import allgather

    heap = get_heap()
    world = heap.world
    if overlap is None:
        overlap = heap.backend == "hip" and world > 1
    if overlap and heap.backend == "hip" and world > 1:
        return _tp_moe_forward_overlapped(x_shard, topk_ids, topk_w,
                                          w_gate_up, w_down, ag_ctx,
                                          coll_ctx)
    x_full = allgather(x_shard, ag_ctx) if world > 1 else x_shard
    return tp_moe_from_full(x_full, topk_ids, topk_w, w_gate_up, w_down,
                            coll_ctx)


def segment_sort_meta(topk_ids_seg: torch.Tensor, n_experts: int,
                      dev, bm: int = 32) -> dict:
    """Expert-sort descriptors for ONE gathered segment, built purely
    from the replicated router output (no activation data needed):
    expert_base/rows, pq work items, the sorted row order, and each
    sorted row's source token within the segment."""
    k = topk_ids_seg.shape[1]
    flat = topk_ids_seg.reshape(-1).to(torch.int64)
    order = torch.argsort(flat, stable=True)
    counts = torch.bincount(flat, minlength=n_experts)
    base = torch.cumsum(counts, 0) - counts
    cnt = counts.cpu().tolist()
    items = [e * 65536 + t for e, c in enumerate(cnt)
             for t in range((c + bm - 1) // bm)]
    return dict(
        expert_base=base.to(torch.int32).to(dev).contiguous(),
        expert_rows=counts.to(torch.int32).to(dev).contiguous(),
        work_items=torch.tensor(items or [0], dtype=torch.int32,
                                device=dev),
        work_count=torch.tensor([len(items)], dtype=torch.int32,
                                device=dev),
        order=order, tok=order // k)


def _tp_moe_forward_overlapped(x_shard, topk_ids, topk_w, w_gate_up,
                               w_down, ag_ctx, coll_ctx):
    """Rank-staggered AG + grouped-GEMM overlap — the reference's AG-MoE
    threadblock-swizzle capability (threadblock_swizzle_ag_moe.cu:248
    rank-staggered tile visit order after AG — behavior only), as an
    MI355X segment-progressive design: the shard push is the standard
    SDMA producer over the comm-stream pool; the expert compute runs in
    arrival-preference order (own shard first, then (rank+1)%world, ...)
    with each segment gated by one single-workgroup flag-wait kernel on
    the compute stream. Grouped GEMMs of arrived segments overlap the
    xGMI transfer of later ones; the router is replicated, so every
    segment's expert-sort metadata is built before any data lands. The
    only spin is the 1-WG wait (peer flags come from SDMA, no CU
    dependency), so this is safe even when ranks share one GPU."""
    from .collectives import reduce_scatter
    from .fused import swiglu_op
    from .. import _C

    heap = get_heap()
    world, rank = heap.world, heap.rank
    m, H = x_shard.shape
    E = w_gate_up.shape[0]
    inter = w_down.shape[2]
    ctx = ag_ctx
    chunks = ctx.chunks_per_rank
    assert H == ctx.k and m <= ctx.max_m_per_rank and m % chunks == 0
    dev = x_shard.device
    compute = torch.cuda.current_stream()
    s = compute.cuda_stream
    # push phase: the allgather SDMA producer, WITHOUT the global wait
    _C.reset_flags(ctx.flags.ptr(), world * chunks, 0, s)
    heap.barrier_all_on_stream(compute)
    seg_bytes = ctx.max_m_per_rank * H * 2
    _C.memcpy_async(ctx.ws.ptr() + rank * seg_bytes, x_shard.data_ptr(),
                    m * H * 2, s)
    _C.reset_flags(ctx.flags.ptr() + rank * chunks * 4, chunks, 1, s)
    ctx.ready_ev.record(compute)
    ns = len(ctx.comm_streams)
    chunk_bytes = (m // chunks) * H * 2
    for st in range(ns):
        ctx.comm_streams[st].wait_event(ctx.ready_ev)
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        stream = ctx.comm_streams[i % ns]
        dst_seg = ctx.ws.ptr(peer) + rank * seg_bytes
        dst_flag = ctx.flags.ptr(peer) + rank * chunks * 4
        for c in range(chunks):
            _C.memcpy_async(dst_seg + c * chunk_bytes,
                            x_shard.data_ptr() + c * chunk_bytes,
                            chunk_bytes, stream.cuda_stream)
            _C.memcpy_async(dst_flag + c * 4, heap.one_src.ptr(), 4,
                            stream.cuda_stream)
    # per-segment sort metadata from the replicated router (no data dep)
    K = topk_ids.shape[1]
    metas = [segment_sort_meta(topk_ids[src * m:(src + 1) * m], E, dev)
             for src in range(world)]
    # segment-progressive expert compute
    ws_view = ctx.ws.local()  # [world, max_m, H]
    y_full = torch.zeros(world * m, H, dtype=torch.float32, device=dev)
    mk = m * K
    for i in range(world):
        src = (rank + i) % world
        if src != rank:  # own segment is compute-stream-ordered already
            _C.wait_eq(ctx.flags.ptr() + src * chunks * 4, chunks, 1, s)
        meta = metas[src]
        x_sorted = torch.zeros(mk + 128, H, dtype=x_shard.dtype,
                               device=dev)
        x_sorted[:mk] = ws_view[src, :m].index_select(0, meta["tok"])
        hidden = grouped_gemm(x_sorted, w_gate_up, meta)
        act = swiglu_op(hidden, inter)
        part = grouped_gemm(act, w_down, meta)
        w_sorted = topk_w[src * m:(src + 1) * m].reshape(-1)[
            meta["order"]].to(torch.float32)
        y_full.index_add_(0, src * m + meta["tok"],
                          part[:mk].float() * w_sorted[:, None])
    for st in range(min(ns, max(world - 1, 1))):
        ctx.join_evs[st].record(ctx.comm_streams[st])
        compute.wait_event(ctx.join_evs[st])
    return reduce_scatter(y_full.to(x_shard.dtype), coll_ctx)


def tp_moe_ref(x_full: torch.Tensor, topk_ids: torch.Tensor,
               topk_w: torch.Tensor, w_gate_up_full: torch.Tensor,
               w_down_full: torch.Tensor, world: int,
               rank: int) -> torch.Tensor:
    """Golden single-process reference with FULL (unsharded) weights:
    x_full [M, H] (the gathered batch), w_gate_up_full [E, 2*inter, H]
    ([gate; up]), w_down_full [E, H, inter]. Returns rank's [M/world, H]
    segment of the full MoE output."""
    import torch.nn.functional as F

    inter = w_down_full.shape[2]
    y = torch.zeros(x_full.shape[0], x_full.shape[1], dtype=torch.float32)
    for t in range(x_full.shape[0]):
        for j in range(topk_ids.shape[1]):
            e = int(topk_ids[t, j])
            hv = x_full[t].float() @ w_gate_up_full[e].float().t()
            a = F.silu(hv[:inter]) * hv[inter:]
            y[t] += float(topk_w[t, j]) * (a @ w_down_full[e].float().t())
    m_local = x_full.shape[0] // world
    lo = rank * m_local
    return y[lo:lo + m_local].to(x_full.dtype)
