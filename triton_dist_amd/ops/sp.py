"""Sequence-parallel ops: distributed flash-decode (seq-sharded KV with
cross-rank LSE merge), Ulysses head<->sequence all-to-all, and ring-AG
prefill attention with online LSE merging.

Capability parity (behavior only) with Triton-distributed:
  kernels/nvidia/flash_decode.py:130-1132 + layers/nvidia/
  sp_flash_decode_layer.py:44-149   — split-KV decode + inter-rank combine
  kernels/nvidia/sp_ulysess_*.py, ulysses_sp_dispatch.py:470-606 — Ulysses
  kernels/nvidia/sp_ag_attention_intra_node.py:106-428 — AG-attention
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch

from ..runtime import cpu_shm
from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap


# ---------------------------------------------------------------------------
# Distributed flash-decode
# ---------------------------------------------------------------------------
@dataclass
class SPFlashDecodeContext:
    heap: SymmHeap
    max_batch: int
    qh: int
    parts: SymmBuffer   # [world, B, qh, 128] fp32
    lses: SymmBuffer    # [world, B, qh] fp32
    flags: SymmBuffer   # [world] int32
    epoch: int = 0

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_sp_flash_decode_context(max_batch: int, qh: int,
                                   heap: Optional[SymmHeap] = None
                                   ) -> SPFlashDecodeContext:
    heap = heap or get_heap()
    w = heap.world
    return SPFlashDecodeContext(
        heap, max_batch, qh,
        parts=heap.alloc_buffer((w, max_batch, qh, 128), torch.float32),
        lses=heap.alloc_buffer((w, max_batch, qh), torch.float32),
        flags=heap.alloc_buffer((w,), torch.int32),
    )


def sp_flash_decode(q: torch.Tensor, kv_k: torch.Tensor, kv_v: torch.Tensor,
                    chunk_len: torch.Tensor, ctx: SPFlashDecodeContext,
                    qh: int, kvh: int,
                    out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Sequence-parallel GQA decode: every rank holds a KV chunk of length
    `chunk_len` (device int64) for ALL sequences; q [B, qh*128] replicated.
    Each rank computes its split-KV partial + LSE, pushes both to every
    peer, and the combine kernel merges. Returns [B, qh*128] bf16 on every
    rank."""
    b = q.shape[0]
    world, rank = ctx.world, ctx.rank
    assert b <= ctx.max_batch and qh == ctx.qh
    heap = ctx.heap
    if heap.backend == "cpu":
        return _sp_flash_decode_cpu(q, kv_k, kv_v, chunk_len, ctx, qh, kvh,
                                    out)
    _C = heap._C
    stream = torch.cuda.current_stream()
    s = stream.cuda_stream
    if out is None:
        out = torch.empty_like(q)

    _C.reset_flags(ctx.flags.ptr(), world, 0, s)
    heap.barrier_all_on_stream(stream)

    part_row = ctx.max_batch * qh * 128 * 4   # bytes per rank slot
    lse_row = ctx.max_batch * qh * 4
    my_part = ctx.parts.ptr() + rank * part_row
    my_lse = ctx.lses.ptr() + rank * lse_row
    _C.flash_decode_partial(q.data_ptr(), kv_k.data_ptr(), kv_v.data_ptr(),
                            my_part, my_lse, chunk_len.data_ptr(), b, qh,
                            kvh, kv_k.shape[1], s)
    # push my partial+lse to peers (SDMA), then my own flag
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        _C.memcpy_async(ctx.parts.ptr(peer) + rank * part_row, my_part,
                        b * qh * 128 * 4, s)
        _C.memcpy_async(ctx.lses.ptr(peer) + rank * lse_row, my_lse,
                        b * qh * 4, s)
        _C.memcpy_async(ctx.flags.ptr(peer) + rank * 4, heap.one_src.ptr(),
                        4, s)
    _C.reset_flags(ctx.flags.ptr() + rank * 4, 1, 1, s)
    # combine waits all flags; slot stride = max_batch, so any
    # b <= max_batch indexes correctly
    _C.lse_combine(ctx.parts.ptr(), ctx.lses.ptr(), out.data_ptr(),
                   ctx.flags.ptr(), world, b, qh, ctx.max_batch, s)
    return out


def _sp_flash_decode_cpu(q, kv_k, kv_v, chunk_len, ctx, qh, kvh, out):
    """CPU mock: torch partial attention with explicit LSE + shm exchange."""
    b = q.shape[0]
    d = 128
    world, rank = ctx.world, ctx.rank
    L = int(chunk_len)
    ctx.epoch += 1
    heap = ctx.heap
    heap.barrier_all()
    g = qh // kvh
    qv = q.view(b, kvh, g, d).float()
    ks = kv_k[:, :L].float()  # [b, L, kvh, d]
    vs = kv_v[:, :L].float()
    scores = torch.einsum("bhgd,blhd->bhgl", qv, ks) / (d ** 0.5)
    m = scores.amax(-1, keepdim=True)
    p = torch.exp(scores - m)
    l = p.sum(-1, keepdim=True)
    o = torch.einsum("bhgl,blhd->bhgd", p / l, vs)
    lse = (m + torch.log(l)).squeeze(-1)  # [b, kvh, g]
    for peer in range(world):
        ctx.parts.peer(peer)[rank, :b].copy_(
            o.reshape(b, qh, d))
        ctx.lses.peer(peer)[rank, :b].copy_(lse.reshape(b, qh))
        cpu_shm.notify(ctx.flags.peer(peer), rank, ctx.epoch)
    fl = ctx.flags.local()
    for r in range(world):
        cpu_shm.wait_ge(fl, r, ctx.epoch)
    parts = ctx.parts.local()[:, :b]   # [world, b, qh, d]
    lses = ctx.lses.local()[:, :b]     # [world, b, qh]
    mx = lses.amax(0)
    w = torch.exp(lses - mx)
    w = w / w.sum(0)
    merged = (parts * w.unsqueeze(-1)).sum(0)
    res = merged.reshape(b, qh * d).to(q.dtype)
    if out is not None:
        out.copy_(res)
        return out
    return res


def sp_flash_decode_ref(q, kv_k_full, kv_v_full, total_len, qh, kvh):
    """Golden: sdpa over the full (gathered) KV."""
    import torch.nn.functional as F

    b = q.shape[0]
    qs = q.view(b, qh, 1, 128).float()
    ks = kv_k_full[:, :total_len].transpose(1, 2).float()
    vs = kv_v_full[:, :total_len].transpose(1, 2).float()
    o = F.scaled_dot_product_attention(qs, ks, vs, enable_gqa=True)
    return o.view(b, qh * 128).to(q.dtype)


# ---------------------------------------------------------------------------
# Ulysses all-to-all (tokens <-> heads resharding)
# ---------------------------------------------------------------------------
@dataclass
class UlyssesContext:
    heap: SymmHeap
    max_tokens: int  # per-rank tokens
    h_loc: int       # heads per rank after resharding
    head_dim: int
    recv: SymmBuffer  # [world, max_tokens, h_loc, D]
    flags: SymmBuffer
    epoch: int = 0

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_ulysses_context(max_tokens: int, n_heads: int, head_dim: int,
                           heap: Optional[SymmHeap] = None) -> UlyssesContext:
    heap = heap or get_heap()
    w = heap.world
    assert n_heads % w == 0
    h_loc = n_heads // w
    return UlyssesContext(
        heap, max_tokens, h_loc, head_dim,
        recv=heap.alloc_buffer((w, max_tokens, h_loc, head_dim),
                               torch.bfloat16),
        flags=heap.alloc_buffer((w,), torch.int32),
    )


def ulysses_a2a(x: torch.Tensor, ctx: UlyssesContext) -> torch.Tensor:
    """[T_loc, n_heads, D] (seq-sharded, all heads) ->
    [world*T_loc, h_loc, D] (all seq, head shard). The inverse resharding is
    the same op with heads<->tokens roles swapped by the caller's reshape.
    """
    t_loc, n_heads, d = x.shape
    world, rank = ctx.world, ctx.rank
    h_loc = ctx.h_loc
    assert n_heads == h_loc * world and d == ctx.head_dim
    assert t_loc <= ctx.max_tokens
    heap = ctx.heap

    if heap.backend == "cpu":
        ctx.epoch += 1
        heap.barrier_all()
        for peer in range(world):
            seg = x[:, peer * h_loc:(peer + 1) * h_loc].contiguous()
            ctx.recv.peer(peer)[rank, :t_loc].copy_(seg)
            cpu_shm.notify(ctx.flags.peer(peer), rank, ctx.epoch)
        fl = ctx.flags.local()
        for r in range(world):
            cpu_shm.wait_ge(fl, r, ctx.epoch)
        return ctx.recv.local()[:, :t_loc].reshape(world * t_loc, h_loc, d) \
            .clone()

    _C = heap._C
    stream = torch.cuda.current_stream()
    s = stream.cuda_stream
    _C.reset_flags(ctx.flags.ptr(), world, 0, s)
    heap.barrier_all_on_stream(stream)
    # contiguous per-peer head slices, then one SDMA per peer + flag
    seg_bytes = t_loc * h_loc * d * 2
    slot_bytes = ctx.max_tokens * h_loc * d * 2
    # send buffer in [peer][t][h_loc][d] order (receiver's layout)
    xs = x.view(t_loc, world, h_loc, d).permute(1, 0, 2, 3).contiguous()
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        _C.memcpy_async(ctx.recv.ptr(peer) + rank * slot_bytes,
                        xs[peer].data_ptr(), seg_bytes, s)
        _C.memcpy_async(ctx.flags.ptr(peer) + rank * 4, heap.one_src.ptr(),
                        4, s)
    _C.memcpy_async(ctx.recv.ptr() + rank * slot_bytes,
                    xs[rank].data_ptr(), seg_bytes, s)
    _C.reset_flags(ctx.flags.ptr() + rank * 4, 1, 1, s)
    _C.wait_eq(ctx.flags.ptr(), world, 1, s)
    return ctx.recv.local()[:, :t_loc].reshape(world * t_loc, h_loc, d)


def ulysses_a2a_ref(x: torch.Tensor, group=None) -> torch.Tensor:
    """Golden: torch.distributed all_to_all of head slices (CPU-staged)."""
    import torch.distributed as dist

    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    t_loc, n_heads, d = x.shape
    h_loc = n_heads // world
    xs = x.view(t_loc, world, h_loc, d).permute(1, 0, 2, 3).contiguous().cpu()
    # a2a via all_gather (gloo has no alltoall): every rank gathers all send
    # matrices, then picks the column addressed to it.
    gathered = [torch.empty_like(xs) for _ in range(world)]
    dist.all_gather(gathered, xs, group=group)
    outs = [gathered[src][rank] for src in range(world)]
    return torch.cat(outs, 0).to(x.device)


# ---------------------------------------------------------------------------
# Ring-AG prefill attention: K/V chunks are all-gathered peer-by-peer on
# comm streams while the consumer runs chunked flash attention with online
# LSE merging (torch aten flash returns the logsumexp on ROCm/aotriton).
# ---------------------------------------------------------------------------
@dataclass
class SPAGAttnContext:
    heap: SymmHeap
    max_tokens: int  # per-rank sequence chunk
    kvh: int
    head_dim: int
    kbuf: SymmBuffer   # [world, B? folded] — [world, max_tokens, kvh, D]
    vbuf: SymmBuffer
    flags: SymmBuffer
    comm_streams: List = field(default_factory=list)
    ready_ev: Optional[object] = None
    join_evs: List = field(default_factory=list)

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank


def create_sp_ag_attn_context(max_chunk_tokens: int, kvh: int, head_dim: int,
                              heap: Optional[SymmHeap] = None
                              ) -> SPAGAttnContext:
    heap = heap or get_heap()
    w = heap.world
    ctx = SPAGAttnContext(
        heap, max_chunk_tokens, kvh, head_dim,
        kbuf=heap.alloc_buffer((w, max_chunk_tokens, kvh, head_dim),
                               torch.bfloat16),
        vbuf=heap.alloc_buffer((w, max_chunk_tokens, kvh, head_dim),
                               torch.bfloat16),
        flags=heap.alloc_buffer((w,), torch.int32),
    )
    if heap.backend == "hip":
        n = max(w - 1, 1)
        ctx.comm_streams = [torch.cuda.Stream() for _ in range(min(n, 7))]
        ctx.ready_ev = torch.cuda.Event()
        ctx.join_evs = [torch.cuda.Event() for _ in ctx.comm_streams]
    return ctx


def _flash_with_lse(q, k, v, causal):
    """q/k/v: [B, S, H, D] bf16 -> (out [B,S,H,D], lse [B,S,H] fp32) via
    the in-house MFMA FA2 kernel (csrc/kernels/attention.hip
    k_flash_prefill); aten flash remains only for head_dim != 128."""
    if q.is_cuda and q.shape[-1] == 128 and q.shape[1] <= 1024:
        from .fused import flash_prefill_op

        return flash_prefill_op(q, k, v, causal=causal, return_lse=True)
    qt = q.permute(0, 2, 1, 3).contiguous()
    kt = k.permute(0, 2, 1, 3).contiguous()
    vt = v.permute(0, 2, 1, 3).contiguous()
    outs = torch.ops.aten._scaled_dot_product_flash_attention(
        qt, kt, vt, dropout_p=0.0, is_causal=causal)
    return outs[0].permute(0, 2, 1, 3), outs[1].permute(0, 2, 1)


def _merge_lse(o1, l1, o2, l2):
    """Merge two attention partials with their logsumexps."""
    m = torch.maximum(l1, l2)
    w1 = torch.exp(l1 - m)
    w2 = torch.exp(l2 - m)
    denom = w1 + w2
    o = (o1 * (w1 / denom).unsqueeze(-1).to(o1.dtype)
         + o2 * (w2 / denom).unsqueeze(-1).to(o2.dtype))
    return o, m + torch.log(denom)


def sp_ag_attention(q: torch.Tensor, k_chunk: torch.Tensor,
                    v_chunk: torch.Tensor, ctx: SPAGAttnContext,
                    qh: int) -> torch.Tensor:
    """Causal self-attention with sequence sharding (zig-zag-free v1:
    contiguous chunks; rank r's queries attend chunks 0..r).

    q: [S_loc, qh, D] (this rank's query chunk, post-RoPE)
    k_chunk/v_chunk: [S_loc, kvh, D] (this rank's KV chunk, post-RoPE)
    Returns [S_loc, qh, D].
    """
    s_loc = q.shape[0]
    world, rank = ctx.world, ctx.rank
    kvh, d = ctx.kvh, ctx.head_dim
    heap = ctx.heap
    assert s_loc <= ctx.max_tokens

    if heap.backend == "cpu":
        return _sp_ag_attention_cpu(q, k_chunk, v_chunk, ctx, qh)

    _C = heap._C
    compute = torch.cuda.current_stream()
    s = compute.cuda_stream
    _C.reset_flags(ctx.flags.ptr(), world, 0, s)
    heap.barrier_all_on_stream(compute)
    slot = ctx.max_tokens * kvh * d * 2
    nbytes = s_loc * kvh * d * 2
    kc = k_chunk.contiguous()
    vc = v_chunk.contiguous()
    # local chunk into my slot + flag
    _C.memcpy_async(ctx.kbuf.ptr() + rank * slot, kc.data_ptr(), nbytes, s)
    _C.memcpy_async(ctx.vbuf.ptr() + rank * slot, vc.data_ptr(), nbytes, s)
    _C.reset_flags(ctx.flags.ptr() + rank * 4, 1, 1, s)
    # push to peers on comm streams (cp-engine producer)
    ctx.ready_ev.record(compute)
    ns = len(ctx.comm_streams)
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        st = ctx.comm_streams[i % ns]
        if i < ns:
            st.wait_event(ctx.ready_ev)
        _C.memcpy_async(ctx.kbuf.ptr(peer) + rank * slot, kc.data_ptr(),
                        nbytes, st.cuda_stream)
        _C.memcpy_async(ctx.vbuf.ptr(peer) + rank * slot, vc.data_ptr(),
                        nbytes, st.cuda_stream)
        _C.memcpy_async(ctx.flags.ptr(peer) + rank * 4, heap.one_src.ptr(),
                        4, st.cuda_stream)

    # consumer: chunks 0..rank (causal), waiting each chunk's flag;
    # the FA2 consumer takes the natural [1, S, H, D] layout directly
    qt = q.unsqueeze(0)  # [1, S_loc, qh, D]
    o = lse = None
    for src in range(rank + 1):
        _C.wait_eq(ctx.flags.ptr() + src * 4, 1, 1, s)
        kv_k = ctx.kbuf.local()[src, :s_loc].unsqueeze(0)
        kv_v = ctx.vbuf.local()[src, :s_loc].unsqueeze(0)
        causal = (src == rank)
        oc, lc = _flash_with_lse(qt, kv_k, kv_v, causal)
        if o is None:
            o, lse = oc, lc
        else:
            o, lse = _merge_lse(o, lse, oc, lc)
    for i, ev in enumerate(ctx.join_evs):
        ev.record(ctx.comm_streams[i])
        compute.wait_event(ev)
    return o.squeeze(0).contiguous()


def _sp_ag_attention_cpu(q, k_chunk, v_chunk, ctx, qh):
    import torch.nn.functional as F

    s_loc = q.shape[0]
    world, rank = ctx.world, ctx.rank
    ctx.epoch = getattr(ctx, "epoch", 0) + 1
    heap = ctx.heap
    heap.barrier_all()
    for peer in range(world):
        ctx.kbuf.peer(peer)[rank, :s_loc].copy_(k_chunk)
        ctx.vbuf.peer(peer)[rank, :s_loc].copy_(v_chunk)
        cpu_shm.notify(ctx.flags.peer(peer), rank, ctx.epoch)
    fl = ctx.flags.local()
    for r in range(world):
        cpu_shm.wait_ge(fl, r, ctx.epoch)
    # full causal attention over gathered prefix (CPU golden-ish path)
    ks = ctx.kbuf.local()[:rank + 1, :s_loc].reshape((rank + 1) * s_loc, -1,
                                                     ctx.head_dim)
    vs = ctx.vbuf.local()[:rank + 1, :s_loc].reshape((rank + 1) * s_loc, -1,
                                                     ctx.head_dim)
    qt = q.permute(1, 0, 2).unsqueeze(0).float()
    kt = ks.permute(1, 0, 2).unsqueeze(0).float()
    vt = vs.permute(1, 0, 2).unsqueeze(0).float()
    total = (rank + 1) * s_loc
    qpos = rank * s_loc + torch.arange(s_loc)
    mask = torch.arange(total).unsqueeze(0) <= qpos.unsqueeze(1)
    o = F.scaled_dot_product_attention(qt, kt, vt,
                                       attn_mask=mask.view(1, 1, s_loc, total),
                                       enable_gqa=True)
    return o.squeeze(0).permute(1, 0, 2).to(q.dtype)


# ---------------------------------------------------------------------------
# Fused Ulysses: qkv-GEMM whose epilogue IS the head all-to-all, and the
# inverse a2a whose consumer is an arrival-ordered accumulating o-GEMM
# (capability parity with the reference's fused pair,
# kernels/nvidia/sp_ulysess_qkv_gemm_all2all.py:64-545 and
# sp_ulysess_o_all2all_gemm.py — behavior only).
# ---------------------------------------------------------------------------
@dataclass
class UlyssesFusedContext:
    heap: SymmHeap
    max_tokens: int      # per-rank tokens
    qkv_dim: int         # (qh + 2*kvh) * head_dim (full, pre-shard)
    o_in_dim: int        # qh_local_after_a2a... = qkv q-part per rank
    recv_qkv: SymmBuffer  # [world_src, max_tokens, qkv_dim/world] bf16
    qkv_flags: SymmBuffer  # [world] int32
    recv_o: SymmBuffer   # [world_src, max_tokens, o_in_dim] bf16
    o_flags: SymmBuffer  # [world] int32
    arrive: Optional[torch.Tensor] = None  # local [world] int32

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank

    @property
    def peer_cols(self):
        return self.qkv_dim // self.world


def create_ulysses_fused_context(max_tokens: int, qkv_dim: int,
                                 o_in_dim: int,
                                 heap: Optional[SymmHeap] = None
                                 ) -> UlyssesFusedContext:
    heap = heap or get_heap()
    w = heap.world
    assert qkv_dim % (w * 256) == 0, "peer column block must tile by 256"
    ctx = UlyssesFusedContext(
        heap, max_tokens, qkv_dim, o_in_dim,
        recv_qkv=heap.alloc_buffer((w, max_tokens, qkv_dim // w),
                                   torch.bfloat16),
        qkv_flags=heap.alloc_buffer((w,), torch.int32),
        recv_o=heap.alloc_buffer((w, max_tokens, o_in_dim), torch.bfloat16),
        o_flags=heap.alloc_buffer((w,), torch.int32),
    )
    if heap.backend == "hip":
        ctx.arrive = torch.zeros(w, dtype=torch.int32, device="cuda")
    return ctx


def ulysses_qkv_gemm_a2a(x: torch.Tensor, w_qkv: torch.Tensor,
                         ctx: UlyssesFusedContext) -> torch.Tensor:
    """[T_loc, H] @ w_qkv[qkv_dim, H]^T fused with the head a2a: the GEMM
    epilogue pushes each output tile to the rank owning its column block.
    Returns the gathered [world * T_loc, qkv_dim/world] view (all tokens,
    my head shard)."""
    t_loc, hdim = x.shape
    world, rank = ctx.world, ctx.rank
    assert w_qkv.shape[0] == ctx.qkv_dim and t_loc <= ctx.max_tokens
    heap = ctx.heap
    pc = ctx.peer_cols

    if heap.backend == "cpu":
        full = (x.float() @ w_qkv.float().t()).to(x.dtype)
        heap.barrier_all()
        for peer in range(world):
            seg = full[:, peer * pc:(peer + 1) * pc]
            ctx.recv_qkv.peer(peer)[rank, :t_loc].copy_(seg)
        heap.barrier_all()
        res = ctx.recv_qkv.local()[:, :t_loc].reshape(world * t_loc, pc)
        out = res.clone()
        heap.barrier_all()
        return out

    _C = heap._C
    stream = torch.cuda.current_stream()
    s = stream.cuda_stream
    assert t_loc % 256 == 0 and hdim % 128 == 0
    _C.reset_flags(ctx.qkv_flags.ptr(), world, 0, s)
    _C.reset_flags(ctx.arrive.data_ptr(), world, 0, s)
    heap.barrier_all_on_stream(stream)
    tiles_per_peer = (t_loc // 256) * (pc // 256)
    _C.gemm256_colscatter(x.data_ptr(), w_qkv.data_ptr(), t_loc,
                          ctx.qkv_dim, hdim, ctx.recv_qkv.offset,
                          ctx.qkv_flags.offset, pc, ctx.max_tokens,
                          ctx.arrive.data_ptr(), tiles_per_peer, 1, s)
    _C.wait_eq(ctx.qkv_flags.ptr(), world, 1, s)
    if t_loc == ctx.max_tokens:
        return ctx.recv_qkv.local().reshape(world * t_loc, pc)
    return ctx.recv_qkv.local()[:, :t_loc].reshape(world * t_loc, pc)


def ulysses_a2a_o_gemm(attn: torch.Tensor, w_o_split: torch.Tensor,
                       ctx: UlyssesFusedContext,
                       out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Inverse a2a fused with the o-projection: attn [world*T_loc, o_in]
    (all tokens, my head shard) is pushed back per-destination (contiguous
    row blocks, SDMA), and the o-GEMM accumulates per-SOURCE K-block
    partials into an fp32 ws as each source's segment arrives (fixed
    visit order, per-src flag waits) — overlap between late arrivals and
    early partial GEMMs. w_o_split: [world, N, o_in] (w_o's K blocks,
    pre-sliced contiguous at layer init)."""
    world, rank = ctx.world, ctx.rank
    t_total = attn.shape[0]
    assert t_total % world == 0
    t_loc = t_total // world
    o_in = ctx.o_in_dim
    n = w_o_split.shape[1]
    heap = ctx.heap

    if heap.backend == "cpu":
        heap.barrier_all()
        for peer in range(world):
            seg = attn[peer * t_loc:(peer + 1) * t_loc]
            ctx.recv_o.peer(peer)[rank, :t_loc].copy_(seg)
        heap.barrier_all()
        acc = torch.zeros(t_loc, n, dtype=torch.float32)
        for src in range(world):
            a_p = ctx.recv_o.local()[src, :t_loc].float()
            acc += a_p @ w_o_split[src].float().t()
        res = acc.to(attn.dtype)
        heap.barrier_all()
        if out is not None:
            out.copy_(res)
            return out
        return res

    _C = heap._C
    stream = torch.cuda.current_stream()
    s = stream.cuda_stream
    assert attn.dtype == torch.bfloat16 and attn.is_contiguous()
    assert t_loc % 256 == 0 and n % 256 == 0 and o_in % 128 == 0
    _C.reset_flags(ctx.o_flags.ptr(), world, 0, s)
    heap.barrier_all_on_stream(stream)
    slot = ctx.max_tokens * o_in * 2
    nbytes = t_loc * o_in * 2
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        _C.memcpy_async(ctx.recv_o.ptr(peer) + rank * slot,
                        attn.data_ptr() + peer * nbytes, nbytes, s)
        _C.memcpy_async(ctx.o_flags.ptr(peer) + rank * 4,
                        heap.one_src.ptr(), 4, s)
    _C.memcpy_async(ctx.recv_o.ptr() + rank * slot,
                    attn.data_ptr() + rank * nbytes, nbytes, s)
    _C.reset_flags(ctx.o_flags.ptr() + rank * 4, 1, 1, s)
    ws = torch.zeros(t_loc, n, dtype=torch.float32, device=attn.device)
    for j in range(world):
        src = (rank + j) % world  # self first (already local)
        _C.wait_eq(ctx.o_flags.ptr() + src * 4, 1, 1, s)
        _C.gemm256_acc_bf16(ctx.recv_o.ptr() + src * slot,
                            w_o_split[src].data_ptr(), ws.data_ptr(),
                            t_loc, n, o_in, s)
    if out is None:
        out = torch.empty(t_loc, n, dtype=torch.bfloat16,
                          device=attn.device)
    _C.f32_to_bf16(ws.data_ptr(), out.data_ptr(), 0, t_loc, n, s)
    return out


def ulysses_fused_ref(x: torch.Tensor, w_qkv: torch.Tensor,
                      w_o: torch.Tensor, group=None):
    """Golden for the fused pair composed: qkv-GEMM -> head a2a ->
    (identity "attention") -> inverse a2a -> o-GEMM, via torch
    collectives on CPU."""
    import torch.distributed as dist

    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    t_loc = x.shape[0]
    full = (x.float() @ w_qkv.float().t())
    pc = w_qkv.shape[0] // world
    # a2a: my shard of every rank's tokens
    segs = [full[:, p * pc:(p + 1) * pc].contiguous().cpu() for p in
            range(world)]
    gathered = [torch.empty_like(segs[0]) for _ in range(world)]
    outs = []
    for p in range(world):
        dist.all_gather(gathered, segs[p], group=group)
        if p == rank:
            outs = [g.clone() for g in gathered]
    mine = torch.cat(outs, 0)  # [world*t_loc, pc]
    return mine.to(x.device)


def sp_ag_attention_zigzag(q: torch.Tensor, k_chunk: torch.Tensor,
                           v_chunk: torch.Tensor, ctx: SPAGAttnContext,
                           qh: int) -> torch.Tensor:
    """Zig-zag-balanced causal SP attention (the reference's causal
    load-balance capability, sp_ag_attention_intra_node.py:283 — behavior
    only). The sequence is pre-sharded into 2*world blocks; rank r holds
    blocks (r, 2*world-1-r) so every rank's attention work is the
    constant 2*world+1 block-pairs instead of r+1.

    q/k_chunk/v_chunk: [2, S_blk, heads, D] (this rank's two blocks,
    ordered [block r, block 2W-1-r], post-RoPE). Returns [2, S_blk, qh,
    D]. Requires 2*S_blk <= ctx.max_tokens (both blocks ride one slot).
    """
    two, s_blk, _, d = q.shape
    assert two == 2 and d == ctx.head_dim
    world, rank = ctx.world, ctx.rank
    kvh = ctx.kvh
    heap = ctx.heap
    assert 2 * s_blk <= ctx.max_tokens

    if heap.backend == "cpu":
        return _sp_ag_attention_zigzag_cpu(q, k_chunk, v_chunk, ctx, qh)

    _C = heap._C
    compute = torch.cuda.current_stream()
    s = compute.cuda_stream
    _C.reset_flags(ctx.flags.ptr(), world, 0, s)
    heap.barrier_all_on_stream(compute)
    slot = ctx.max_tokens * kvh * d * 2
    nbytes = 2 * s_blk * kvh * d * 2
    kc = k_chunk.contiguous()
    vc = v_chunk.contiguous()
    _C.memcpy_async(ctx.kbuf.ptr() + rank * slot, kc.data_ptr(), nbytes, s)
    _C.memcpy_async(ctx.vbuf.ptr() + rank * slot, vc.data_ptr(), nbytes, s)
    _C.reset_flags(ctx.flags.ptr() + rank * 4, 1, 1, s)
    ctx.ready_ev.record(compute)
    ns = len(ctx.comm_streams)
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        st = ctx.comm_streams[i % ns]
        if i < ns:
            st.wait_event(ctx.ready_ev)
        _C.memcpy_async(ctx.kbuf.ptr(peer) + rank * slot, kc.data_ptr(),
                        nbytes, st.cuda_stream)
        _C.memcpy_async(ctx.vbuf.ptr(peer) + rank * slot, vc.data_ptr(),
                        nbytes, st.cuda_stream)
        _C.memcpy_async(ctx.flags.ptr(peer) + rank * 4, heap.one_src.ptr(),
                        4, st.cuda_stream)

    def kv_half(src, half):
        base = ctx.kbuf.local()[src, :2 * s_blk].reshape(2, s_blk, kvh, d)
        vb = ctx.vbuf.local()[src, :2 * s_blk].reshape(2, s_blk, kvh, d)
        return base[half].unsqueeze(0), vb[half].unsqueeze(0)

    outs = []
    for half in range(2):
        # q block's global index: half 0 -> rank; half 1 -> 2W-1-rank
        qt = q[half].unsqueeze(0)
        o = lse = None
        # sources in (src, their half, causal?) order
        plan = []
        if half == 0:
            for j in range(rank):
                plan.append((j, 0, False))
            plan.append((rank, 0, True))
        else:
            for j in range(world):
                plan.append((j, 0, False))       # every A block
            for j in range(rank + 1, world):
                plan.append((j, 1, False))       # B blocks below mine
            plan.append((rank, 1, True))
        for src, kv_h, causal in plan:
            _C.wait_eq(ctx.flags.ptr() + src * 4, 1, 1, s)
            kk, vv = kv_half(src, kv_h)
            oc, lc = _flash_with_lse(qt, kk, vv, causal)
            o, lse = (oc, lc) if o is None else _merge_lse(o, lse, oc, lc)
        outs.append(o.squeeze(0))
    for i, ev in enumerate(ctx.join_evs):
        ev.record(ctx.comm_streams[i])
        compute.wait_event(ev)
    return torch.stack(outs, 0).contiguous()


def _sp_ag_attention_zigzag_cpu(q, k_chunk, v_chunk, ctx, qh):
    import torch.nn.functional as F

    two, s_blk, kvh, d = k_chunk.shape
    world, rank = ctx.world, ctx.rank
    heap = ctx.heap
    ctx.epoch = getattr(ctx, "epoch", 0) + 1
    heap.barrier_all()
    flat_k = k_chunk.reshape(2 * s_blk, kvh, d)
    flat_v = v_chunk.reshape(2 * s_blk, kvh, d)
    for peer in range(world):
        ctx.kbuf.peer(peer)[rank, :2 * s_blk].copy_(flat_k)
        ctx.vbuf.peer(peer)[rank, :2 * s_blk].copy_(flat_v)
        cpu_shm.notify(ctx.flags.peer(peer), rank, ctx.epoch)
    fl = ctx.flags.local()
    for r in range(world):
        cpu_shm.wait_ge(fl, r, ctx.epoch)
    # reassemble the FULL zig-zag sequence in global block order
    nb = 2 * world
    blocks_k = [None] * nb
    blocks_v = [None] * nb
    for src in range(world):
        kb = ctx.kbuf.local()[src, :2 * s_blk].reshape(2, s_blk, kvh, d)
        vb = ctx.vbuf.local()[src, :2 * s_blk].reshape(2, s_blk, kvh, d)
        blocks_k[src] = kb[0]
        blocks_k[nb - 1 - src] = kb[1]
        blocks_v[src] = vb[0]
        blocks_v[nb - 1 - src] = vb[1]
    ks = torch.cat(blocks_k, 0)
    vs = torch.cat(blocks_v, 0)
    total = nb * s_blk
    outs = []
    for half, gidx in ((0, rank), (1, nb - 1 - rank)):
        qt = q[half].permute(1, 0, 2).unsqueeze(0).float()
        kt = ks.permute(1, 0, 2).unsqueeze(0).float()
        vt = vs.permute(1, 0, 2).unsqueeze(0).float()
        qpos = gidx * s_blk + torch.arange(s_blk)
        mask = torch.arange(total).unsqueeze(0) <= qpos.unsqueeze(1)
        o = F.scaled_dot_product_attention(
            qt, kt, vt, attn_mask=mask.view(1, 1, s_blk, total),
            enable_gqa=True)
        outs.append(o.squeeze(0).permute(1, 0, 2).to(q.dtype))
    return torch.stack(outs, 0)
