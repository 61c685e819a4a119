"""Expert-parallel MoE ops: token dispatch -> grouped expert GEMM (SwiGLU)
-> combine, over the symmetric heap.

Capability parity with the reference's EP stack (Triton-distributed
layers/amd/ep_a2a_layer.py:208-548 EPAll2AllLayer dispatch/combine,
kernels/amd/ep_a2a_intra_node.py, function/amd/ep_moe_fused.py fused_ep_moe
— behavior only; see csrc/kernels/moe.hip for the MI355X-native design:
deterministic splits-based layout, expert-sorted recv, per-source
completion signals).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch

from ..runtime import cpu_shm
from ..runtime.symm_mem import SymmBuffer, SymmHeap, get_heap
from ..utils.distributed import gpu_oversubscribed


@dataclass
class EPContext:
    heap: SymmHeap
    max_tokens: int     # per-rank tokens per call
    hidden: int
    n_experts: int
    topk: int
    cap: int            # recv rows capacity = world * max_tokens * topk
    # symmetric buffers
    all_splits: SymmBuffer      # [world, E] int32
    splits_flags: SymmBuffer    # [world] int32
    recv_x: SymmBuffer          # [cap + 128, H] bf16 (tail slack for GEMM)
    meta: SymmBuffer            # [cap, 2] int32 (src, tok_k)
    combine_buf: SymmBuffer     # [max_tokens * topk, H] bf16
    disp_flags: SymmBuffer      # [world] int32 (fp8 path: per-dst)
    comb_flags: SymmBuffer      # [world] int32
    eflags: Optional[SymmBuffer] = None  # [world, e_loc] int32 per-expert
    # local device state
    local: dict = field(default_factory=dict)
    epoch: int = 0
    low_latency: bool = False
    credit_flags: Optional[SymmBuffer] = None
    call_no: int = 0
    fp8: bool = False
    recv_q: Optional[SymmBuffer] = None
    recv_scale: Optional[SymmBuffer] = None

    @property
    def world(self):
        return self.heap.world

    @property
    def rank(self):
        return self.heap.rank

    @property
    def e_loc(self):
        return self.n_experts // self.world


def create_ep_context(max_tokens: int, hidden: int, n_experts: int,
                      topk: int, heap: Optional[SymmHeap] = None,
                      low_latency: bool = False,
                      fp8: bool = False) -> EPContext:
    """low_latency=True allocates DOUBLE buffers (call-parity indexed) and
    the op skips the entry barrier + flag resets: flags carry the running
    call number and a credit flag bounds buffer reuse — the reference's LL
    protocol (low_latency_all_to_all.py:36-170 act_pos=call%2 — behavior
    only)."""
    heap = heap or get_heap()
    world = heap.world
    assert n_experts % world == 0
    cap = world * max_tokens * topk
    nbuf = 2 if low_latency else 1
    ctx = EPContext(
        heap=heap, max_tokens=max_tokens, hidden=hidden,
        n_experts=n_experts, topk=topk, cap=cap,
        all_splits=heap.alloc_buffer((nbuf, world, n_experts), torch.int32),
        splits_flags=heap.alloc_buffer((nbuf, world), torch.int32),
        recv_x=heap.alloc_buffer((nbuf, cap + 128, hidden), torch.bfloat16),
        meta=heap.alloc_buffer((nbuf, cap, 2), torch.int32),
        combine_buf=heap.alloc_buffer((nbuf, max_tokens * topk, hidden),
                                      torch.bfloat16),
        disp_flags=heap.alloc_buffer((nbuf, world), torch.int32),
        comb_flags=heap.alloc_buffer((nbuf, world), torch.int32),
    )
    ctx.eflags = heap.alloc_buffer((nbuf, world, n_experts // world),
                                   torch.int32)
    ctx.low_latency = low_latency
    ctx.credit_flags = heap.alloc_buffer((world,), torch.int32) \
        if low_latency else None
    ctx.call_no = 0
    ctx.fp8 = fp8
    if fp8:
        assert hidden % 128 == 0
        ctx.recv_q = heap.alloc_buffer((nbuf, cap, hidden), torch.uint8)
        ctx.recv_scale = heap.alloc_buffer((nbuf, cap, hidden // 128),
                                           torch.float32)
    if heap.backend == "hip":
        dev = "cuda"
        e = n_experts
        ctx.local = dict(
            counts=torch.zeros(e, dtype=torch.int32, device=dev),
            send_pos=torch.zeros(max_tokens * topk, dtype=torch.int32,
                                 device=dev),
            send_base=torch.zeros(e, dtype=torch.int32, device=dev),
            send_to_dst=torch.zeros(world, dtype=torch.int32, device=dev),
            expert_base=torch.zeros(ctx.e_loc, dtype=torch.int32, device=dev),
            expert_rows=torch.zeros(ctx.e_loc, dtype=torch.int32, device=dev),
            recv_from_src=torch.zeros(world, dtype=torch.int32, device=dev),
            recv_total=torch.zeros(1, dtype=torch.int32, device=dev),
            arrive_d=torch.zeros(world, dtype=torch.int32, device=dev),
            arrive_e=torch.zeros(e, dtype=torch.int32, device=dev),
            arrive_c=torch.zeros(world, dtype=torch.int32, device=dev),
            call_cell=torch.zeros(1, dtype=torch.int32, device=dev),
            work_items=torch.zeros(ctx.e_loc + cap // 32 + 1,
                                   dtype=torch.int32, device=dev),
            work_count=torch.zeros(1, dtype=torch.int32, device=dev),
        )
        ctx.local["comm_stream"] = torch.cuda.Stream()
        ctx.local["ev_fork"] = torch.cuda.Event()
        ctx.local["ev_join"] = torch.cuda.Event()
    return ctx


def ep_moe_forward(x: torch.Tensor, topk_ids: torch.Tensor,
                   topk_w: torch.Tensor, w_gate_up: torch.Tensor,
                   w_down: torch.Tensor, ctx: EPContext,
                   out: Optional[torch.Tensor] = None,
                   fused_swiglu: bool = False) -> torch.Tensor:
    """Full EP MoE layer for this rank's tokens.

    x: [T, H] bf16; topk_ids: [T, K] int32 (global expert ids, -1 = drop);
    topk_w: [T, K] fp32; w_gate_up: [E_loc, 2*I, H]; w_down: [E_loc, H, I].
    Returns [T, H] bf16.
    """
    T, H = x.shape
    K = ctx.topk
    E, e_loc = ctx.n_experts, ctx.e_loc
    world, rank = ctx.world, ctx.rank
    assert T <= ctx.max_tokens and H == ctx.hidden
    inter = w_down.shape[2]
    if out is None:
        out = torch.empty(T, H, dtype=torch.bfloat16, device=x.device)

    if ctx.heap.backend == "cpu":
        return _ep_moe_cpu(x, topk_ids, topk_w, w_gate_up, w_down, ctx, out)

    _C = ctx.heap._C
    heap = ctx.heap
    stream = torch.cuda.current_stream()
    s = stream.cuda_stream
    L = ctx.local
    nbuf = 2 if ctx.low_latency else 1
    parity = ctx.call_no % nbuf
    ctx.call_no += 1
    # per-parity buffer offsets
    splits_off = ctx.all_splits.offset + parity * world * E * 4
    sflags_off = ctx.splits_flags.offset + parity * world * 4
    recv_x_off = ctx.recv_x.offset + parity * (ctx.cap + 128) * H * 2
    meta_off = ctx.meta.offset + parity * ctx.cap * 2 * 4
    combine_off = ctx.combine_buf.offset + parity * ctx.max_tokens * K * H * 2
    dflags_off = ctx.disp_flags.offset + parity * world * 4
    cflags_off = ctx.comb_flags.offset + parity * world * 4
    eflags_off = ctx.eflags.offset + parity * world * e_loc * 4
    cell = 0

    # world == 1 fast path: no peers — stream order replaces every
    # reset/flag/barrier kernel (splits "exchange" = reading my own
    # counts; dispatch is just the expert-sort permute; GEMMs ungated).
    # Saves ~10 control-kernel launches per call (ROADMAP item 6).
    if world == 1:
        return _ep_moe_world1(x, topk_ids, topk_w, w_gate_up, w_down, ctx,
                              out, parity, recv_x_off, meta_off,
                              combine_off, cflags_off, eflags_off,
                              fused_swiglu)

    # phase 0: buffer-reuse protection
    L["counts"].zero_()
    L["arrive_d"].zero_()
    L["arrive_e"].zero_()
    L["arrive_c"].zero_()
    if ctx.low_latency:
        # LL protocol: no barrier, no flag resets — flags carry the call
        # number; a credit flag bounds parity-buffer reuse to 2 calls
        cell = L["call_cell"].data_ptr()
        _C.bump_cell(cell, s)
        _C.wait_flags_ge_cell(ctx.credit_flags.ptr(), world, cell, -2, s)
    else:
        _C.reset_flags(heap.ptr(rank, sflags_off), world, 0, s)
        _C.reset_flags(heap.ptr(rank, dflags_off), world, 0, s)
        _C.reset_flags(heap.ptr(rank, cflags_off), world, 0, s)
        _C.reset_flags(heap.ptr(rank, eflags_off), world * e_loc, 0, s)
        heap.barrier_all_on_stream(stream)

    # phase 1: routing histogram + slot assignment
    _C.moe_count(topk_ids.data_ptr(), L["counts"].data_ptr(),
                 L["send_pos"].data_ptr(), L["send_to_dst"].data_ptr(),
                 T * K, E, e_loc, world, s)

    # phase 2: exchange the splits matrix (SDMA push + flag per peer)
    my_row_off = splits_off + rank * E * 4
    _C.memcpy_async(heap.ptr(rank, my_row_off), L["counts"].data_ptr(),
                    E * 4, s)
    if ctx.low_latency:
        _C.memcpy_async(heap.ptr(rank, sflags_off) + rank * 4, cell, 4, s)
    else:
        _C.reset_flags(heap.ptr(rank, sflags_off) + rank * 4, 1, 1, s)
    for i in range(world - 1):
        peer = (rank + 1 + i) % world
        _C.memcpy_async(heap.ptr(peer, my_row_off), L["counts"].data_ptr(),
                        E * 4, s)
        src = cell if ctx.low_latency else heap.one_src.ptr()
        _C.memcpy_async(heap.ptr(peer, sflags_off) + rank * 4, src, 4, s)
    if ctx.low_latency:
        _C.wait_flags_ge_cell(heap.ptr(rank, sflags_off), world, cell, 0, s)
    else:
        _C.wait_eq(heap.ptr(rank, sflags_off), world, 1, s)

    # phase 3: derive layouts (+ the grouped-GEMM work queue)
    avg_rows = max(1, (T * K * world) // E)
    small_m = avg_rows <= 64
    bm = 32 if small_m else 128
    _C.moe_layout(heap.ptr(rank, splits_off), rank, world, E, e_loc,
                  L["send_base"].data_ptr(), L["expert_base"].data_ptr(),
                  L["expert_rows"].data_ptr(), L["recv_from_src"].data_ptr(),
                  L["recv_total"].data_ptr(), s,
                  L["work_items"].data_ptr(), L["work_count"].data_ptr(), bm)

    # phase 4: dispatch (xGMI row push + per-dst completion signals);
    # fp8 mode quantizes the payload (groupwise scales) and dequantizes
    # after the wait — halves the wire bytes
    safe = gpu_oversubscribed(world)
    forked = False
    if ctx.fp8:
        recv_q_off = ctx.recv_q.offset + parity * ctx.cap * H
        recv_s_off = ctx.recv_scale.offset + parity * ctx.cap * (H // 128) * 4
        _C.moe_dispatch_fp8(x.data_ptr(), topk_ids.data_ptr(),
                            L["send_pos"].data_ptr(),
                            L["send_base"].data_ptr(),
                            L["send_to_dst"].data_ptr(), recv_q_off,
                            recv_s_off, meta_off, dflags_off,
                            L["arrive_d"].data_ptr(), T, K, H, e_loc, s,
                            cell)
        _C.moe_wait_flags(heap.ptr(rank, dflags_off), world, s, cell)
        if not (small_m and H % 128 == 0):
            # large-M fallback: standalone dequant + bf16 grouped GEMM
            _C.moe_dequant(heap.ptr(rank, recv_q_off),
                           heap.ptr(rank, recv_s_off),
                           heap.ptr(rank, recv_x_off),
                           L["recv_total"].data_ptr(), ctx.cap, H, s)
    elif safe:
        # ranks sharing one GPU (validation boxes): the spin-gated
        # overlap paths can cross-process starve — both ranks' GEMM
        # grids occupy every CU slot spinning on eflags while the peer
        # process's dispatch kernel waits for a slot, until the 30 s
        # spin watchdog traps. Serialize instead: dispatch on the
        # compute stream, one-workgroup wait on all world*e_loc
        # (src, expert) flags, then UNGATED grouped GEMMs.
        _C.moe_dispatch(x.data_ptr(), topk_ids.data_ptr(),
                        L["send_pos"].data_ptr(), L["send_base"].data_ptr(),
                        L["counts"].data_ptr(), recv_x_off,
                        meta_off, eflags_off,
                        L["arrive_e"].data_ptr(), T, K, H, e_loc, E, s,
                        cell)
        if ctx.low_latency:
            _C.wait_flags_ge_cell(heap.ptr(rank, eflags_off),
                                  world * e_loc, cell, 0, s)
        else:
            _C.wait_eq(heap.ptr(rank, eflags_off), world * e_loc, 1, s)
    elif not (small_m and not ctx.low_latency):
        # dispatch rides the comm stream; the expert GEMM is gated
        # per-(expert, tile) on eflags, so FFN tiles of early-complete
        # experts run while slow sources still stream (per-expert
        # overlap; reference kernels/amd/ep_all2all_fused.py:316
        # capability — behavior only)
        forked = True
        comm = L["comm_stream"]
        L["ev_fork"].record(stream)
        comm.wait_event(L["ev_fork"])
        _C.moe_dispatch(x.data_ptr(), topk_ids.data_ptr(),
                        L["send_pos"].data_ptr(), L["send_base"].data_ptr(),
                        L["counts"].data_ptr(), recv_x_off,
                        meta_off, eflags_off,
                        L["arrive_e"].data_ptr(), T, K, H, e_loc, E,
                        comm.cuda_stream, cell)
        L["ev_join"].record(comm)

    # phase 5: grouped expert FFN — persistent work-queue kernel when
    # experts are lightly loaded (decode: ~T*K*world/E rows per expert)
    cap_tiles = (ctx.cap + bm - 1) // bm
    # +128-row slack everywhere an edge GEMM tile may over-read
    expert_h = None if fused_swiglu else torch.empty(
        ctx.cap + 128, 2 * inter, dtype=torch.bfloat16, device=x.device)
    gate = 0 if (ctx.fp8 or safe) else heap.ptr(rank, eflags_off)
    act = torch.empty(ctx.cap + 128, inter, dtype=torch.bfloat16,
                      device=x.device)
    g1_out = act if fused_swiglu else expert_h
    fs = 1 if fused_swiglu else 0
    if small_m and ctx.fp8 and H % 128 == 0:
        # fp8 A stays on the wire format: per-fragment dequant in
        # registers inside the grouped GEMM (no recv_x round trip)
        recv_q_off2 = ctx.recv_q.offset + parity * ctx.cap * H
        recv_s_off2 = ctx.recv_scale.offset \
            + parity * ctx.cap * (H // 128) * 4
        _C.moe_grouped_gemm_pq_fp8(
            heap.ptr(rank, recv_q_off2), heap.ptr(rank, recv_s_off2),
            w_gate_up.data_ptr(), g1_out.data_ptr(),
            L["expert_base"].data_ptr(), L["expert_rows"].data_ptr(),
            L["work_items"].data_ptr(), L["work_count"].data_ptr(),
            2 * inter, H, fs, s)
    elif small_m and not ctx.fp8 and not ctx.low_latency and not safe:
        # SINGLE-LAUNCH mega-kernel: dispatch producer workgroups +
        # per-expert-gated grouped GEMM in one kernel (closes the
        # reference's ep_all2all_fused.py:316 row completely)
        _C.moe_fused_dispatch_gemm(
            x.data_ptr(), topk_ids.data_ptr(), L["send_pos"].data_ptr(),
            L["send_base"].data_ptr(), L["counts"].data_ptr(), recv_x_off,
            meta_off, eflags_off, L["arrive_e"].data_ptr(), T, K, H,
            e_loc, E, w_gate_up.data_ptr(), g1_out.data_ptr(),
            L["expert_base"].data_ptr(), L["expert_rows"].data_ptr(),
            L["work_items"].data_ptr(), L["work_count"].data_ptr(),
            2 * inter, H, fs, s, cell)
    elif small_m:
        _C.moe_grouped_gemm_pq(heap.ptr(rank, recv_x_off),
                               w_gate_up.data_ptr(), g1_out.data_ptr(),
                               L["expert_base"].data_ptr(),
                               L["expert_rows"].data_ptr(),
                               L["work_items"].data_ptr(),
                               L["work_count"].data_ptr(), 2 * inter, H, s,
                               gate, cell, world, e_loc, fs)
    else:
        _C.moe_grouped_gemm(heap.ptr(rank, recv_x_off), w_gate_up.data_ptr(),
                            g1_out.data_ptr(), L["expert_base"].data_ptr(),
                            L["expert_rows"].data_ptr(), e_loc, cap_tiles,
                            2 * inter, H, ctx.cap, s, False, gate, cell,
                            world, fs)
    if not fused_swiglu:
        _C.swiglu(expert_h.data_ptr(), act.data_ptr(), ctx.cap + 128, inter,
                  s)
    expert_out = torch.empty(ctx.cap + 128, H, dtype=torch.bfloat16,
                             device=x.device)
    if small_m:
        _C.moe_grouped_gemm_pq(act.data_ptr(), w_down.data_ptr(),
                               expert_out.data_ptr(),
                               L["expert_base"].data_ptr(),
                               L["expert_rows"].data_ptr(),
                               L["work_items"].data_ptr(),
                               L["work_count"].data_ptr(), H, inter, s)
    else:
        _C.moe_grouped_gemm(act.data_ptr(), w_down.data_ptr(),
                            expert_out.data_ptr(), L["expert_base"].data_ptr(),
                            L["expert_rows"].data_ptr(), e_loc, cap_tiles, H,
                            inter, ctx.cap, s, False)

    # phase 6: combine (return rows + weighted reduce); join the
    # dispatch fork first (graph hygiene — by now it long completed)
    if forked:
        stream.wait_event(L["ev_join"])
    _C.moe_combine_send(expert_out.data_ptr(), heap.ptr(rank, meta_off),
                        L["recv_total"].data_ptr(),
                        L["recv_from_src"].data_ptr(),
                        combine_off, cflags_off,
                        L["arrive_c"].data_ptr(), ctx.cap, H, s, cell)
    _C.moe_combine_reduce(heap.ptr(rank, combine_off), topk_w.data_ptr(),
                          topk_ids.data_ptr(), out.data_ptr(),
                          heap.ptr(rank, cflags_off), world, T, K, H, E, s,
                          cell)
    if ctx.low_latency:
        _C.signal_credit(ctx.credit_flags.offset, cell, s)
    return out


def _ep_moe_world1(x, topk_ids, topk_w, w_gate_up, w_down, ctx, out,
                   parity, recv_x_off, meta_off, combine_off, cflags_off,
                   eflags_off, fused_swiglu=False):
    T, H = x.shape
    K = ctx.topk
    E, e_loc = ctx.n_experts, ctx.e_loc
    inter = w_down.shape[2]
    _C = ctx.heap._C
    heap = ctx.heap
    rank = 0
    s = torch.cuda.current_stream().cuda_stream
    L = ctx.local
    L["counts"].zero_()
    L["arrive_e"].zero_()
    L["arrive_c"].zero_()
    _C.moe_count(topk_ids.data_ptr(), L["counts"].data_ptr(),
                 L["send_pos"].data_ptr(), L["send_to_dst"].data_ptr(),
                 T * K, E, e_loc, 1, s)
    avg_rows = max(1, T * K // E)
    small_m = avg_rows <= 64
    bm = 32 if small_m else 128
    # splits matrix [1, E] IS my counts tensor — no exchange
    _C.moe_layout(L["counts"].data_ptr(), 0, 1, E, e_loc,
                  L["send_base"].data_ptr(), L["expert_base"].data_ptr(),
                  L["expert_rows"].data_ptr(), L["recv_from_src"].data_ptr(),
                  L["recv_total"].data_ptr(), s,
                  L["work_items"].data_ptr(), L["work_count"].data_ptr(), bm)
    # expert-sort permute (local copies; signals are cheap local stores
    # and nothing waits on them)
    _C.moe_dispatch(x.data_ptr(), topk_ids.data_ptr(),
                    L["send_pos"].data_ptr(), L["send_base"].data_ptr(),
                    L["counts"].data_ptr(), recv_x_off, meta_off,
                    eflags_off, L["arrive_e"].data_ptr(), T, K, H, e_loc,
                    E, s, 0)
    cap_tiles = (ctx.cap + bm - 1) // bm
    act = torch.empty(ctx.cap + 128, inter, dtype=torch.bfloat16,
                      device=x.device)
    expert_h = None if fused_swiglu else torch.empty(
        ctx.cap + 128, 2 * inter, dtype=torch.bfloat16, device=x.device)
    g1_out = act if fused_swiglu else expert_h
    fs = 1 if fused_swiglu else 0
    if small_m:
        _C.moe_grouped_gemm_pq(heap.ptr(rank, recv_x_off),
                               w_gate_up.data_ptr(), g1_out.data_ptr(),
                               L["expert_base"].data_ptr(),
                               L["expert_rows"].data_ptr(),
                               L["work_items"].data_ptr(),
                               L["work_count"].data_ptr(), 2 * inter, H, s,
                               0, 0, 1, e_loc, fs)
    else:
        _C.moe_grouped_gemm(heap.ptr(rank, recv_x_off), w_gate_up.data_ptr(),
                            g1_out.data_ptr(), L["expert_base"].data_ptr(),
                            L["expert_rows"].data_ptr(), e_loc, cap_tiles,
                            2 * inter, H, ctx.cap, s, False, 0, 0, 1, fs)
    if not fused_swiglu:
        _C.swiglu(expert_h.data_ptr(), act.data_ptr(), ctx.cap + 128, inter,
                  s)
    expert_out = torch.empty(ctx.cap + 128, H, dtype=torch.bfloat16,
                             device=x.device)
    if small_m:
        _C.moe_grouped_gemm_pq(act.data_ptr(), w_down.data_ptr(),
                               expert_out.data_ptr(),
                               L["expert_base"].data_ptr(),
                               L["expert_rows"].data_ptr(),
                               L["work_items"].data_ptr(),
                               L["work_count"].data_ptr(), H, inter, s)
    else:
        _C.moe_grouped_gemm(act.data_ptr(), w_down.data_ptr(),
                            expert_out.data_ptr(),
                            L["expert_base"].data_ptr(),
                            L["expert_rows"].data_ptr(), e_loc, cap_tiles,
                            H, inter, ctx.cap, s, False)
    _C.moe_combine_send(expert_out.data_ptr(), heap.ptr(rank, meta_off),
                        L["recv_total"].data_ptr(),
                        L["recv_from_src"].data_ptr(),
                        combine_off, cflags_off,
                        L["arrive_c"].data_ptr(), ctx.cap, H, s, 0)
    _C.moe_combine_reduce(heap.ptr(rank, combine_off), topk_w.data_ptr(),
                          topk_ids.data_ptr(), out.data_ptr(),
                          heap.ptr(rank, cflags_off), 1, T, K, H, E, s, 0)
    return out


def _ep_moe_cpu(x, topk_ids, topk_w, w_gate_up, w_down, ctx, out):
    """CPU mock: same deterministic layout algebra over the shm heap."""
    T, H = x.shape
    K, E, e_loc = ctx.topk, ctx.n_experts, ctx.e_loc
    world, rank = ctx.world, ctx.rank
    inter = w_down.shape[2]
    ctx.epoch += 1
    heap = ctx.heap
    heap.barrier_all()

    ids = topk_ids.reshape(-1)
    counts = torch.zeros(E, dtype=torch.int32)
    send_pos = torch.full((T * K,), -1, dtype=torch.int32)
    for i in range(T * K):
        e = int(ids[i])
        if 0 <= e < E:
            send_pos[i] = counts[e]
            counts[e] += 1
    # splits exchange
    for p in range(world):
        ctx.all_splits.peer(p)[0, rank].copy_(counts)
        cpu_shm.notify(ctx.splits_flags.peer(p)[0], rank, ctx.epoch)
    fl = ctx.splits_flags.local()[0]
    for sidx in range(world):
        cpu_shm.wait_ge(fl, sidx, ctx.epoch)
    splits = ctx.all_splits.local()[0].clone()  # [world, E]

    # layouts (same algebra as k_moe_layout)
    send_base = torch.zeros(E, dtype=torch.int64)
    expert_base = torch.zeros(e_loc, dtype=torch.int64)
    expert_rows = torch.zeros(e_loc, dtype=torch.int64)
    for d in range(world):
        base = 0
        for le in range(e_loc):
            e = d * e_loc + le
            off = 0
            for src in range(world):
                if src == rank:
                    send_base[e] = base + off
                off += int(splits[src, e])
            if d == rank:
                expert_base[le] = base
                expert_rows[le] = off
            base += off

    # dispatch
    for i in range(T * K):
        e = int(ids[i])
        if e < 0 or e >= E or send_pos[i] < 0:
            continue
        d = e // e_loc
        slot = int(send_base[e]) + int(send_pos[i])
        ctx.recv_x.peer(d)[0, slot].copy_(x[i // K])
        ctx.meta.peer(d)[0, slot, 0] = rank
        ctx.meta.peer(d)[0, slot, 1] = i
    heap.barrier_all()

    # grouped FFN
    expert_out = torch.zeros(ctx.cap + 128, H, dtype=torch.bfloat16)
    rx = ctx.recv_x.local()[0]
    for le in range(e_loc):
        nb = int(expert_rows[le])
        if nb == 0:
            continue
        b0 = int(expert_base[le])
        xe = rx[b0:b0 + nb].float()
        h = xe @ w_gate_up[le].float().t()
        a = torch.nn.functional.silu(h[:, :inter]) * h[:, inter:]
        y = a @ w_down[le].float().t()
        expert_out[b0:b0 + nb] = y.to(torch.bfloat16)

    # combine send
    meta = ctx.meta.local()[0]
    total = int(expert_base[-1] + expert_rows[-1]) if e_loc else 0
    for r in range(total):
        src = int(meta[r, 0])
        tok_k = int(meta[r, 1])
        ctx.combine_buf.peer(src)[0, tok_k].copy_(expert_out[r])
    heap.barrier_all()

    # reduce
    cb = ctx.combine_buf.local()[0]
    acc = torch.zeros(T, H, dtype=torch.float32)
    for t in range(T):
        for k in range(K):
            e = int(ids[t * K + k])
            if e < 0 or e >= E:
                continue
            acc[t] += float(topk_w[t, k]) * cb[t * K + k].float()
    out.copy_(acc.to(torch.bfloat16))
    heap.barrier_all()
    return out


def ep_moe_ref(x: torch.Tensor, topk_ids: torch.Tensor, topk_w: torch.Tensor,
               full_w_gate_up: torch.Tensor, full_w_down: torch.Tensor,
               group=None) -> torch.Tensor:
    """Golden single-device reference for THIS rank's tokens: every rank has
    the full expert weights [E, ...]; no communication needed."""
    T, H = x.shape
    E = full_w_gate_up.shape[0]
    inter = full_w_down.shape[2]
    K = topk_ids.shape[1]
    acc = torch.zeros(T, H, dtype=torch.float32, device=x.device)
    for t in range(T):
        for k in range(K):
            e = int(topk_ids[t, k])
            if e < 0 or e >= E:
                continue
            h = x[t].float() @ full_w_gate_up[e].float().t()
            a = torch.nn.functional.silu(h[:inter]) * h[inter:]
            y = a @ full_w_down[e].float().t()
            acc[t] += float(topk_w[t, k]) * y
    return acc.to(x.dtype)
