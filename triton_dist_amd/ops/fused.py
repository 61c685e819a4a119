"""Fused memory-bound ops: HIP kernels on GPU, torch reference on CPU.
On GPU hosts the native extension is REQUIRED — these raise rather than
silently falling back to eager torch."""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F


def _C():
    from .. import _C as mod

    if mod is None:
        raise RuntimeError("triton_dist_amd._C missing on a GPU host")
    return mod


def _s():
    return torch.cuda.current_stream().cuda_stream


def rms_norm_op(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-6
                ) -> torch.Tensor:
    if not x.is_cuda:
        x32 = x.float()
        x32 = x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + eps)
        return (x32 * w.float()).to(x.dtype)
    rows, cols = x.shape[0], x.shape[-1]
    out = torch.empty_like(x)
    _C().rmsnorm(x.data_ptr(), w.data_ptr(), out.data_ptr(),
                 x.numel() // cols, cols, eps, _s())
    return out


def add_rms_norm_op(delta: torch.Tensor, resid: torch.Tensor,
                    w: torch.Tensor, eps: float = 1e-6
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """new_resid = resid + delta; normed = rmsnorm(new_resid) * w."""
    if not delta.is_cuda:
        nr = (resid.float() + delta.float()).to(delta.dtype)
        return nr, rms_norm_op(nr, w, eps)
    cols = delta.shape[-1]
    new_resid = torch.empty_like(delta)
    normed = torch.empty_like(delta)
    _C().add_rmsnorm(delta.data_ptr(), resid.data_ptr(),
                     new_resid.data_ptr(), w.data_ptr(), normed.data_ptr(),
                     delta.numel() // cols, cols, eps, _s())
    return new_resid, normed


def swiglu_op(h: torch.Tensor, inter: int) -> torch.Tensor:
    """h: [M, 2*inter] = [gate | up] -> silu(gate) * up, bf16."""
    if not h.is_cuda:
        g, u = h[:, :inter].float(), h[:, inter:].float()
        return (F.silu(g) * u).to(h.dtype)
    out = torch.empty(h.shape[0], inter, dtype=h.dtype, device=h.device)
    _C().swiglu(h.data_ptr(), out.data_ptr(), h.shape[0], inter, _s())
    return out


def qkv_prologue_decode_op(qkv: torch.Tensor, kv_k: torch.Tensor,
                           kv_v: torch.Tensor, offset: torch.Tensor,
                           cos_t: torch.Tensor, sin_t: torch.Tensor,
                           qnw: torch.Tensor, knw: torch.Tensor,
                           qh: int, kvh: int, eps: float,
                           use_qk_norm: bool) -> torch.Tensor:
    """Fused q/k head-RMSNorm + RoPE + KV-cache append for one decode token.
    qkv: [B, (qh+2kvh)*128]; kv_k/kv_v: [B, max_len, kvh, 128].
    Returns rotated q [B, qh*128]."""
    b = qkv.shape[0]
    max_len = kv_k.shape[1]
    q_out = torch.empty(b, qh * 128, dtype=qkv.dtype, device=qkv.device)
    _C().qkv_prologue_decode(
        qkv.data_ptr(), q_out.data_ptr(), kv_k.data_ptr(), kv_v.data_ptr(),
        cos_t.data_ptr(), sin_t.data_ptr(), qnw.data_ptr(), knw.data_ptr(),
        offset.data_ptr(), b, qh, kvh, max_len, eps, use_qk_norm, _s())
    return q_out


def qkv_prologue_prefill_op(qkv: torch.Tensor, kv_k: torch.Tensor,
                            kv_v: torch.Tensor, cos_t: torch.Tensor,
                            sin_t: torch.Tensor, qnw: torch.Tensor,
                            knw: torch.Tensor, b: int, s: int, qh: int,
                            kvh: int, eps: float,
                            use_qk_norm: bool) -> torch.Tensor:
    """Fused prefill prologue: per-head RMSNorm (optional) + RoPE + KV
    cache fill, straight from the qkv projection. qkv: [b*s, (qh+2kvh)*
    128]; kv_k/kv_v: [b, max_len, kvh, 128]. Returns q [b, s, qh, 128]
    (the FA2-native layout). Replaces the eager torch rmsnorm/rotary/
    transpose/copy chain on the prefill path."""
    max_len = kv_k.shape[1]
    q_out = torch.empty(b, s, qh, 128, dtype=qkv.dtype, device=qkv.device)
    _C().qkv_prologue_prefill(
        qkv.data_ptr(), q_out.data_ptr(), kv_k.data_ptr(), kv_v.data_ptr(),
        cos_t.data_ptr(), sin_t.data_ptr(), qnw.data_ptr(), knw.data_ptr(),
        b, s, qh, kvh, max_len, eps, use_qk_norm, _s())
    return q_out


def flash_prefill_op(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     causal: bool = True, return_lse: bool = False,
                     kb_stride: int = 0):
    """FA2 forward on MFMA (csrc/kernels/attention.hip k_flash_prefill).

    q: [b, s, qh, 128]; k/v: [b, s, kvh, 128] — the layer's natural
    post-RoPE layout (no transposes). Returns [b, s, qh, 128].
    Replaces torch sdpa on the prefill hot path (reference capability:
    kernels/nvidia/sp_ag_attention_intra_node.py:257-428 FA2 consumer —
    behavior only)."""
    b, sq, qh, d = q.shape
    kvh = k.shape[-2]
    assert d == 128
    if kb_stride == 0:
        assert k.shape[1] == sq
    q = q.contiguous()
    if kb_stride == 0:
        k = k.contiguous()
        v = v.contiguous()
    out = torch.empty_like(q)
    scale = 1.0 / (d ** 0.5)
    lse = None
    lse_ptr = 0
    if return_lse:
        lse = torch.empty(b, sq, qh, dtype=torch.float32, device=q.device)
        lse_ptr = lse.data_ptr()
    _C().flash_prefill(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       out.data_ptr(), b, sq, qh, kvh, scale, causal,
                       _s(), lse_ptr, kb_stride)
    if return_lse:
        return out, lse
    return out


def flash_decode_op(q: torch.Tensor, kv_k: torch.Tensor, kv_v: torch.Tensor,
                    offset: torch.Tensor, qh: int, kvh: int) -> torch.Tensor:
    """q: [B, qh*128] post-RoPE -> out [B, qh*128]."""
    b = q.shape[0]
    max_len = kv_k.shape[1]
    out = torch.empty_like(q)
    _C().flash_decode(q.data_ptr(), kv_k.data_ptr(), kv_v.data_ptr(),
                      out.data_ptr(), offset.data_ptr(), b, qh, kvh,
                      max_len, _s())
    return out


def flash_decode_paged_op(q: torch.Tensor, paged_cache, layer: int,
                          offset: torch.Tensor, qh: int,
                          kvh: int) -> torch.Tensor:
    """GQA flash-decode over a PagedKVCache (models/kv_cache.py): the
    kernel indexes the block pool through the per-sequence table
    (EXPERIMENTAL — csrc/kernels/attention.hip k_flash_decode_paged). On
    CPU, gathers to contiguous and reuses the torch reference."""
    b = q.shape[0]
    if not q.is_cuda:
        import torch.nn.functional as F

        upto = int(offset.item()) + 1
        kc, vc = paged_cache.gather_layer(layer, upto)
        d = kc.shape[3]
        qs = q.view(b, qh, 1, d).float()
        ks = kc.transpose(1, 2).float()
        vs = vc.transpose(1, 2).float()
        ref = F.scaled_dot_product_attention(qs, ks, vs, enable_gqa=True)
        return ref.view(b, qh * d).to(q.dtype)
    from .. import _C
    out = torch.empty_like(q)
    _C.flash_decode_paged(
        q.contiguous().data_ptr(),
        paged_cache.k_pool[layer].data_ptr(),
        paged_cache.v_pool[layer].data_ptr(),
        paged_cache.block_table.data_ptr(),
        paged_cache.blocks_per_seq, paged_cache.block,
        out.data_ptr(), offset.data_ptr(), b, qh, kvh,
        torch.cuda.current_stream().cuda_stream)
    return out
