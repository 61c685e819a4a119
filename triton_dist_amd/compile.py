"""torch.compile integration: register the fused HIP ops as torch custom
ops with fake (meta) kernels so Dynamo/Inductor graphs can contain them.

Capability parity with Triton-distributed tools/monkey_inductor.py:49-348
(which monkey-patches Inductor's Triton launcher so triton_dist kernels
survive torch.compile) — here the ops are plain custom ops, so no patching
is needed: Inductor treats them as opaque calls and fuses around them.
"""
from __future__ import annotations

import torch

_REGISTERED = False


def register_custom_ops():
    """Idempotently register td::* custom ops (rms_norm, add_rms_norm,
    swiglu, flash_decode)."""
    global _REGISTERED
    if _REGISTERED:
        return
    from .ops import fused

    @torch.library.custom_op("td::rms_norm", mutates_args=())
    def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float
                 ) -> torch.Tensor:
        return fused.rms_norm_op(x.contiguous(), w, eps)

    @rms_norm.register_fake
    def _(x, w, eps):
        return torch.empty_like(x)

    @torch.library.custom_op("td::swiglu", mutates_args=())
    def swiglu(h: torch.Tensor, inter: int) -> torch.Tensor:
        return fused.swiglu_op(h.contiguous(), inter)

    @swiglu.register_fake
    def _(h, inter):
        return h.new_empty(h.shape[0], inter)

    @torch.library.custom_op("td::flash_decode", mutates_args=())
    def flash_decode(q: torch.Tensor, kv_k: torch.Tensor,
                     kv_v: torch.Tensor, offset: torch.Tensor, qh: int,
                     kvh: int) -> torch.Tensor:
        return fused.flash_decode_op(q.contiguous(), kv_k, kv_v, offset,
                                     qh, kvh)

    @flash_decode.register_fake
    def _(q, kv_k, kv_v, offset, qh, kvh):
        return torch.empty_like(q)

    @torch.library.custom_op("td::add_rms_norm", mutates_args=())
    def add_rms_norm(x: torch.Tensor, resid: torch.Tensor, w: torch.Tensor,
                     eps: float) -> tuple[torch.Tensor, torch.Tensor]:
        return fused.add_rms_norm_op(x.contiguous(), resid.contiguous(), w,
                                     eps)

    @add_rms_norm.register_fake
    def _(x, resid, w, eps):
        return torch.empty_like(x), torch.empty_like(x)

    @torch.library.custom_op("td::gdn_decode", mutates_args=("state",))
    def gdn_decode(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   g: torch.Tensor, beta: torch.Tensor, scale: float,
                   state: torch.Tensor) -> torch.Tensor:
        from .ops.gdn import gdn_decode_step
        return gdn_decode_step(q, k, v, g, beta, scale, state)

    @gdn_decode.register_fake
    def _(q, k, v, g, beta, scale, state):
        return q.new_empty(q.shape[0], q.shape[1], v.shape[-1])

    _REGISTERED = True
    return torch.ops.td
