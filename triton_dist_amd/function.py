"""Functional API (capability parity with Triton-distributed
python/triton_dist/function/amd/ep_moe_fused.py:48-97: `fused_ep_moe` and a
torch.autograd.Function wrapper whose backward raises — the framework is
inference-only, same as the reference)."""
from __future__ import annotations

import torch

from .ops.ep_moe import EPContext, ep_moe_forward


def fused_ep_moe(x: torch.Tensor, topk_ids: torch.Tensor,
                 topk_w: torch.Tensor, w_gate_up: torch.Tensor,
                 w_down: torch.Tensor, ctx: EPContext) -> torch.Tensor:
    """dispatch -> grouped GEMM-1 -> SwiGLU -> grouped GEMM-2 -> combine,
    one fused EP pipeline call."""
    return ep_moe_forward(x, topk_ids, topk_w, w_gate_up, w_down, ctx)


class FusedEpMoEFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx_ag, x, topk_ids, topk_w, w_gate_up, w_down, ep_ctx):
        return fused_ep_moe(x, topk_ids, topk_w, w_gate_up, w_down, ep_ctx)

    @staticmethod
    def backward(ctx_ag, grad_out):
        raise NotImplementedError(
            "triton_dist_amd is inference-only (reference parity: "
            "function/amd/ep_moe_fused.py backward also raises)")


def fused_tp_moe(x_shard: torch.Tensor, topk_ids: torch.Tensor,
                 topk_w: torch.Tensor, w_gate_up: torch.Tensor,
                 w_down: torch.Tensor, ag_ctx, coll_ctx) -> torch.Tensor:
    """TP-MoE functional form: AG -> sorted grouped GEMMs -> topk reduce
    -> reduce_scatter (see ops/moe_tp.py)."""
    from .ops.moe_tp import tp_moe_forward

    return tp_moe_forward(x_shard, topk_ids, topk_w, w_gate_up, w_down,
                          ag_ctx, coll_ctx)
