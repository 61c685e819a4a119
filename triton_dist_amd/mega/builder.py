"""Megakernel task-graph builder + levelized round-robin scheduler.

Capability parity with the reference megakernel core (Triton-distributed
mega_triton_kernel/core/{graph.py:59-134 Graph/Node, builder.py:48 task
descriptors, scheduler.py:103-157 static round-robin scheduling} — the
reference codegens a Triton kernel; here descriptors feed the fixed HIP
task vocabulary in csrc/kernels/megakernel.hip).

Task record layout (must match mk::Task):
  int32 x 6: type, score_slot, dep0, dep0_n, dep1, dep1_n  (+8B pad)
  int64 x 13: args (device pointers / scalars)
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Tuple

import torch

# task types (mirror mk::TaskType)
T_RMSNORM = 0
T_ADD_RMSNORM = 1
T_GEMM_TILE = 2
T_SWIGLU = 3
T_QKV_PROLOGUE = 4
T_FLASH_DECODE = 5
T_EMBED = 6
T_KV_ADVANCE = 7
T_GEMM_TILE_PART = 8   # K-range partial -> fp32 ws slice
T_TILE_REDUCE = 9      # sum ws slices -> bf16 C tile

TASK_INT64S = 13
TASK_WORDS = 2 + 1 + TASK_INT64S  # 6 int32 = 3 int64 words (with pad)


@dataclass
class Op:
    """One logical op = a scoreboard slot; its tasks arrive-count to it."""
    slot: int
    level: int
    n_tasks: int = 0


@dataclass
class MegaGraph:
    tasks: List[Tuple] = field(default_factory=list)  # (type, slot, deps, args)
    ops: List[Op] = field(default_factory=list)
    level: int = 0

    def new_op(self) -> Op:
        op = Op(slot=len(self.ops), level=self.level)
        self.ops.append(op)
        return op

    def next_level(self):
        self.level += 1

    def set_level(self, lvl: int):
        """Explicit level placement (pipelined chains interleave group
        levels; the scheduler orders queues by task level, so any
        monotone-with-deps assignment keeps the deadlock-freedom
        argument intact)."""
        self.level = lvl

    def add_task(self, ttype: int, op: Op, args: List[int],
                 deps: List[Tuple[Op, int]] = ()):
        d = [(-1, 0), (-1, 0)]
        for i, (dop, need) in enumerate(deps):
            d[i] = (dop.slot, need if need else dop.n_tasks)
        assert len(args) <= TASK_INT64S
        self.tasks.append((ttype, op.slot, d[0][0], d[0][1], d[1][0],
                           d[1][1], list(args) + [0] * (TASK_INT64S - len(args)),
                           self.level))
        op.n_tasks += 1

    # ---------------------------------------------------------------- build
    def finalize(self, n_wg: int, device="cuda"):
        """Levelized round-robin assignment -> device descriptor tensors."""
        # encode tasks
        n = len(self.tasks)
        buf = torch.zeros(n, TASK_WORDS, dtype=torch.int64)
        for i, (tt, slot, d0, d0n, d1, d1n, args, _lvl) in enumerate(self.tasks):
            # pack int32 pairs into int64 words (little-endian)
            buf[i, 0] = (tt & 0xFFFFFFFF) | ((slot & 0xFFFFFFFF) << 32)
            buf[i, 1] = (d0 & 0xFFFFFFFF) | ((d0n & 0xFFFFFFFF) << 32)
            buf[i, 2] = (d1 & 0xFFFFFFFF) | ((d1n & 0xFFFFFFFF) << 32)
            for j, a in enumerate(args):
                buf[i, 3 + j] = a
        # levelized round-robin queues
        order = sorted(range(n), key=lambda i: self.tasks[i][7])
        queues = [[] for _ in range(n_wg)]
        for pos, ti in enumerate(order):
            queues[pos % n_wg].append(ti)
        flat, offs = [], [0]
        for q in queues:
            flat.extend(q)
            offs.append(len(flat))
        return (buf.to(device),
                torch.tensor(flat, dtype=torch.int32, device=device),
                torch.tensor(offs, dtype=torch.int32, device=device),
                torch.zeros(len(self.ops), dtype=torch.int32, device=device))


T_GEMV = 10            # bsz<=4 decode GEMV (512-col chunk x full K)


def emit_gemv(g: "MegaGraph", a_ptr: int, w_ptr: int, c_ptr: int,
              batch: int, n: int, k: int, dep):
    """Decode-GEMV emission (batch <= 4): one task per 512-column chunk
    streaming the full K — ~25x fewer tasks than 32x128 tiling at bsz 1,
    which is what the ~0.25 us/task dispatch overhead demands. x rows
    must fit the 40 KiB task LDS (m*k*2 <= 40960)."""
    assert batch * k * 2 <= 40960
    chunk = 512
    while n % chunk:
        chunk //= 2
    op = g.new_op()
    for c0 in range(0, n, chunk):
        g.add_task(T_GEMV, op,
                   [a_ptr, w_ptr, c_ptr, batch, n, k, c0, chunk],
                   [(dep, 0)] if dep else [])
    g.next_level()
    return op


T_PRO_FLASH_DECODE = 11  # hop fusion: qkv prologue slice + flash decode
T_GEMM_TILE_PART_NR = 12  # hop fusion: K-partial with on-the-fly rmsnorm A


def emit_gemm(g: "MegaGraph", a_ptr: int, w_ptr: int, c_ptr: int,
              batch: int, n: int, k: int, dep, ksplit: int = 1,
              ws_ptr: int = 0, norm=None, dep2=None):
    """Emit one logical GEMM as scoreboard ops. ksplit == 1: one
    T_GEMM_TILE per 32x128 C tile (the original scheme). ksplit > 1
    (TD_MK_KSPLIT, opt-in): each tile becomes `ksplit` K-range partial
    tasks writing fp32 ws slices plus one reduce task — shortens the
    per-op critical path from a full-K tile to K/ksplit (the 59 ms vs
    15 ms megakernel gap is op-chain latency; docs/ROADMAP.md #4).
    norm=(lnw_ptr, res_ptr) (TD_MK_FUSE, ksplit>1 only): the A operand
    is rmsnorm(a [+ res]) computed inside each partial — the
    one-task-per-row norm hop disappears from the critical path. dep2:
    optional second dependency (the ping-pong residual update).
    Returns the op consumers must depend on."""
    tiles_m = (batch + 31) // 32
    tiles_n = n // 128
    deps = ([(dep, 0)] if dep else []) + ([(dep2, 0)] if dep2 else [])
    if ksplit <= 1 or k % (64 * ksplit):
        assert norm is None, "norm fusion requires the ksplit path"
        op = g.new_op()
        for pm in range(tiles_m):
            for pn in range(tiles_n):
                g.add_task(T_GEMM_TILE, op,
                           [a_ptr, w_ptr, c_ptr, batch, n, k, pm, pn],
                           deps)
        g.next_level()
        return op
    assert ws_ptr, "ksplit > 1 needs an fp32 ws [ksplit, batch_pad, n]"
    klen = k // ksplit
    parts = g.new_op()
    for pm in range(tiles_m):
        for pn in range(tiles_n):
            for sk in range(ksplit):
                if norm is None:
                    g.add_task(T_GEMM_TILE_PART, parts,
                               [a_ptr, w_ptr, ws_ptr, batch, n, k, pm, pn,
                                sk * klen, klen, sk], deps)
                else:
                    g.add_task(T_GEMM_TILE_PART_NR, parts,
                               [a_ptr, w_ptr, ws_ptr, batch, n, k, pm, pn,
                                sk * klen, klen, sk, norm[0], norm[1]],
                               deps)
    g.next_level()
    red = g.new_op()
    for pm in range(tiles_m):
        for pn in range(tiles_n):
            g.add_task(T_TILE_REDUCE, red,
                       [ws_ptr, c_ptr, batch, n, pm, pn, ksplit],
                       [(parts, 0)])
    g.next_level()
    return red


class MegaRun:
    """Owns the finalized descriptors; one launch per decode step."""

    def __init__(self, graph: MegaGraph, n_wg: int = 768, device="cuda"):
        self.n_wg = n_wg
        (self.task_buf, self.queue, self.queue_off,
         self.scoreboard) = graph.finalize(n_wg, device)
        self.n_ops = len(graph.ops)

    def launch(self, stream=None):
        import os

        from .. import _C

        s = stream or torch.cuda.current_stream()
        self.scoreboard.zero_()
        prof_ptr = 0
        if os.environ.get("TD_MK_PROF"):
            if not hasattr(self, "prof"):
                self.prof = torch.zeros(16, 2, dtype=torch.int64,
                                        device="cuda")
            prof_ptr = self.prof.data_ptr()
        _C.megakernel(self.task_buf.data_ptr(), self.queue.data_ptr(),
                      self.queue_off.data_ptr(), self.scoreboard.data_ptr(),
                      self.n_wg, s.cuda_stream,
                      int(os.environ.get("TD_MK_FENCE", "0")), prof_ptr)
