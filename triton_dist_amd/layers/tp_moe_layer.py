"""Tensor-parallel MoE layer: every rank holds an intermediate-dim shard
of EVERY expert; tokens are gathered, expert-sorted and grouped-GEMMed,
and the partial outputs are topk-reduced then reduce-scattered.

Capability parity (behavior only) with the reference's MoE-TP pair
(Triton-distributed kernels/nvidia/allgather_group_gemm.py +
moe_reduce_rs.py and models/qwen_moe.py TP_MoE), the counterpart of
EPMoELayer — the model picks between them (reference: env `EP_MOE`;
here: ModelConfig.moe_impl / env `TD_MOE_IMPL`).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ..ops.moe_tp import tp_moe_from_full
from ..runtime.symm_mem import SymmHeap, get_heap
from .ep_moe_layer import EPMoELayer


class TPMoELayer(EPMoELayer):
    """Shares the router with EPMoELayer; re-shards the expert weights by
    the intermediate dimension instead of by expert."""

    def __init__(self, hidden: int, moe_inter: int, n_experts: int,
                 topk: int, norm_topk: bool = True,
                 heap: Optional[SymmHeap] = None, device="cpu",
                 dtype=torch.bfloat16):
        self.heap = heap or get_heap()
        self.world, self.rank = self.heap.world, self.heap.rank
        assert moe_inter % self.world == 0
        self.hidden, self.inter = hidden, moe_inter
        self.inter_shard_moe = moe_inter // self.world
        self.n_experts, self.topk = n_experts, topk
        self.e_loc = n_experts  # every rank sees every expert
        self.norm_topk = norm_topk
        self.device, self.dtype = device, dtype
        self.router = torch.empty(n_experts, hidden, device=device,
                                  dtype=dtype)
        self.w_gate_up = torch.empty(n_experts, 2 * self.inter_shard_moe,
                                     hidden, device=device, dtype=dtype)
        self.w_down = torch.empty(n_experts, hidden, self.inter_shard_moe,
                                  device=device, dtype=dtype)
        self.ag_ctx = None
        self.coll_ctx = None

    def init_ctx(self, max_tokens: int, ctx=None):
        """max_tokens = the LOCAL token shard size. `ctx` optionally
        shares another TPMoELayer's (ag_ctx, coll_ctx) pair."""
        if ctx is None:
            from ..ops.allgather_gemm import create_ag_gemm_context
            from ..ops.collectives import create_coll_context
            ag = create_ag_gemm_context(max_m_per_rank=max_tokens,
                                        k=self.hidden, heap=self.heap)
            coll = create_coll_context(
                max_seg_elems=max_tokens * self.hidden, heap=self.heap)
            ctx = (ag, coll)
        self.ag_ctx, self.coll_ctx = ctx
        return ctx

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [T_local, H] token shard -> [T_local, H]."""
        from ..ops.allgather_gemm import allgather

        x_full = allgather(x, self.ag_ctx) if self.world > 1 else x
        topk_ids, topk_w = self.route(x_full)
        return tp_moe_from_full(x_full, topk_ids, topk_w, self.w_gate_up,
                                self.w_down, self.coll_ctx)

    __call__ = forward

    def torch_fwd(self, x: torch.Tensor) -> torch.Tensor:
        """Golden reference for REPLICATED x [M, H]: my intermediate-shard
        partial for every token, then all-reduce of the partials."""
        import torch.nn.functional as F

        topk_ids, topk_w = self.route(x)
        m = x.shape[0]
        acc = torch.zeros(m, self.hidden, dtype=torch.float32,
                          device=x.device)
        i_s = self.inter_shard_moe
        for e in range(self.n_experts):
            sel = (topk_ids == e)
            if not sel.any():
                continue
            tok, kk = sel.nonzero(as_tuple=True)
            xe = x[tok].float()
            h = xe @ self.w_gate_up[e].float().t()
            a = F.silu(h[:, :i_s]) * h[:, i_s:]
            contrib = a @ self.w_down[e].float().t()
            acc.index_add_(0, tok,
                           contrib * topk_w[tok, kk].float()[:, None])
        if self.world > 1 and dist.is_initialized():
            dist.all_reduce(acc)
        return acc.to(x.dtype)
