"""Tensor-parallel attention (GQA + RoPE + q/k norm, Qwen3-style) with the
reference's TP modes (Triton-distributed python/triton_dist/layers/amd/
tp_attn.py:171-303 — capability parity, MI355X-native ops):

  ag_rs     — batch-sharded tokens: AG-GEMM(qkv) -> attention on local head
              shard over ALL tokens -> GEMM-RS(o)
  allreduce — replicated: local GEMMs + RCCL all-reduce after o-proj
  torch     — eager golden reference

Attention itself uses torch sdpa (ROCm AOTriton flash backend) over the
static KV cache with a device-offset validity mask, so the decode step is
hipGraph-capturable end to end.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..ops import (ag_gemm, create_ag_gemm_context, create_allreduce_context,
                   create_gemm_rs_context, gemm, gemm_allreduce, gemm_rs)
from ..runtime.symm_mem import SymmHeap, get_heap
from .norm import Rotary, rms_norm


class TP_Attn:
    def __init__(self, hidden: int, n_heads: int, n_kv_heads: int,
                 head_dim: int, rotary: Rotary, mode: str = "ag_rs",
                 qk_norm: bool = True, heap: Optional[SymmHeap] = None,
                 device="cpu", dtype=torch.bfloat16, rms_eps: float = 1e-6):
        self.heap = heap or get_heap()
        self.world, self.rank = self.heap.world, self.heap.rank
        assert n_heads % self.world == 0 and n_kv_heads % self.world == 0
        self.hidden, self.head_dim = hidden, head_dim
        self.qh = n_heads // self.world
        self.kvh = n_kv_heads // self.world
        self.mode = mode
        self.rotary = rotary
        self.qk_norm = qk_norm
        self.rms_eps = rms_eps
        self.device, self.dtype = device, dtype
        qkv_dim = (self.qh + 2 * self.kvh) * head_dim
        self.w_qkv = torch.empty(qkv_dim, hidden, device=device, dtype=dtype)
        self.w_o = torch.empty(hidden, self.qh * head_dim, device=device,
                               dtype=dtype)
        self.q_norm_w = torch.ones(head_dim, device=device, dtype=dtype)
        self.k_norm_w = torch.ones(head_dim, device=device, dtype=dtype)
        self.ag_ctx = None
        self.rs_ctx = None
        self.ar_ctx = None

    def init_weights(self, std=0.02, seed: Optional[int] = None):
        g = None
        if seed is not None:
            g = torch.Generator(device=self.device).manual_seed(seed)
        for w in (self.w_qkv, self.w_o):
            tmp = torch.randn(w.shape, generator=g, device=self.device,
                              dtype=torch.float32) * std
            w.copy_(tmp.to(self.dtype))

    def init_ctx(self, max_m_total: int, ag_ctx=None, rs_ctx=None):
        if self.mode == "gemm_ar":
            if ag_ctx is None:
                ag_ctx = create_allreduce_context(max_m_total * self.hidden,
                                                  heap=self.heap)
            self.ar_ctx = ag_ctx
            return ag_ctx, None
        if self.mode != "ag_rs":
            return None, None
        if ag_ctx is None:
            ag_ctx = create_ag_gemm_context(max_m_total // self.world,
                                            self.hidden, heap=self.heap)
        if rs_ctx is None:
            rs_ctx = create_gemm_rs_context(max_m_total, self.hidden,
                                            heap=self.heap)
        self.ag_ctx, self.rs_ctx = ag_ctx, rs_ctx
        return ag_ctx, rs_ctx

    # ------------------------------------------------------------------ core
    def _qkv_split(self, qkv: torch.Tensor, b: int, s: int):
        d = self.head_dim
        q = qkv[:, :self.qh * d].view(b, s, self.qh, d)
        k = qkv[:, self.qh * d:(self.qh + self.kvh) * d].view(b, s, self.kvh, d)
        v = qkv[:, (self.qh + self.kvh) * d:].view(b, s, self.kvh, d)
        return q, k, v

    def _attention(self, qkv: torch.Tensor, kv_cache, layer_idx: int,
                   pos: torch.Tensor, b: int, s: int, prefill: bool,
                   native: bool = False) -> torch.Tensor:
        """qkv: [b*s, (qh+2kvh)*D] -> attention output [b*s, qh*D]."""
        if (native and not prefill and s == 1 and qkv.is_cuda
                and self.head_dim == 128 and kv_cache is not None
                and self.qh % self.kvh == 0 and self.qh // self.kvh <= 16):
            # fused HIP decode path: qk-norm + RoPE + cache append, then
            # GQA flash-decode (csrc/kernels/{elementwise,attention}.hip)
            from ..ops.fused import flash_decode_op, qkv_prologue_decode_op

            q_rot = qkv_prologue_decode_op(
                qkv, kv_cache.k[layer_idx], kv_cache.v[layer_idx],
                kv_cache.offset, self.rotary.cos, self.rotary.sin,
                self.q_norm_w, self.k_norm_w, self.qh, self.kvh,
                self.rms_eps, self.qk_norm)
            return flash_decode_op(q_rot, kv_cache.k[layer_idx],
                                   kv_cache.v[layer_idx], kv_cache.offset,
                                   self.qh, self.kvh)
        # fully fused prefill prologue + FA2 (fresh-prefill cache fill):
        # one kernel does qk-norm + RoPE + cache fill; FA2 reads K/V from
        # the cache through the batch stride — the torch rmsnorm/rotary/
        # transpose/copy chain is gone from the hot path
        if (qkv.is_cuda and self.head_dim == 128 and prefill
                and kv_cache is not None and s > 1 and s <= 1024):
            from ..ops.fused import flash_prefill_op, qkv_prologue_prefill_op

            kc = kv_cache.k[layer_idx]
            vc = kv_cache.v[layer_idx]
            q4 = qkv_prologue_prefill_op(
                qkv.to(self.dtype), kc, vc, self.rotary.cos,
                self.rotary.sin, self.q_norm_w, self.k_norm_w, b, s,
                self.qh, self.kvh, self.rms_eps, self.qk_norm)
            kb = kc.shape[1] * self.kvh * self.head_dim
            o = flash_prefill_op(q4, kc, vc, causal=True, kb_stride=kb)
            return o.reshape(b * s, self.qh * self.head_dim)
        q, k, v = self._qkv_split(qkv, b, s)
        if self.qk_norm:
            q = rms_norm(q, self.q_norm_w, self.rms_eps)
            k = rms_norm(k, self.k_norm_w, self.rms_eps)
        q, k = self.rotary.apply(q, k, pos)
        # prefill attention: the in-house MFMA FA2 kernel consumes the
        # natural [b, s, h, D] layout directly (no transposes); sdpa
        # remains only for CPU / non-128 head dims / masked decode
        # FA2 measured 2.2-2.8x faster than sdpa at prefill s <= ~512
        # (88.8 vs 198.8 us at b=32 s=128); sdpa/aotriton still wins at
        # long context (s >= 2k) until the kernel pipelines KV tiles —
        # route by sequence length (profiles/README.md r02)
        use_fa2 = (q.is_cuda and self.head_dim == 128 and s <= 1024
                   and (kv_cache is None or prefill))
        if use_fa2:
            from ..ops.fused import flash_prefill_op

            if prefill and kv_cache is not None:
                kv_cache.fill_prefill(layer_idx, k, v)
            o = flash_prefill_op(q.to(self.dtype), k.to(self.dtype),
                                 v.to(self.dtype), causal=(s > 1))
            return o.reshape(b * s, self.qh * self.head_dim)
        q = q.transpose(1, 2)  # [b, qh, s, D]
        if kv_cache is None:
            ks, vs = k.transpose(1, 2), v.transpose(1, 2)
            o = F.scaled_dot_product_attention(q, ks, vs, is_causal=(s > 1),
                                               enable_gqa=True)
        elif prefill:
            kv_cache.fill_prefill(layer_idx, k, v)
            ks, vs = k.transpose(1, 2), v.transpose(1, 2)
            o = F.scaled_dot_product_attention(q, ks, vs, is_causal=(s > 1),
                                               enable_gqa=True)
        else:
            kv_cache.append(layer_idx, k, v)
            ks, vs, mask = kv_cache.view(layer_idx, s)
            o = F.scaled_dot_product_attention(q, ks, vs, attn_mask=mask,
                                               enable_gqa=True)
        return o.transpose(1, 2).reshape(b * s, self.qh * self.head_dim) \
            .contiguous()

    def forward(self, x: torch.Tensor, kv_cache=None, layer_idx: int = 0,
                pos: Optional[torch.Tensor] = None, b: int = 1, s: int = 1,
                prefill: bool = False) -> torch.Tensor:
        if self.mode == "ag_rs":
            qkv = ag_gemm(x, self.w_qkv, self.ag_ctx)       # [M, qkv_dim]
            attn = self._attention(qkv, kv_cache, layer_idx, pos, b, s,
                                   prefill, native=True).to(self.dtype)
            return gemm_rs(attn, self.w_o, self.rs_ctx)     # [M/world, H]
        if self.mode == "gemm_ar":
            qkv = gemm(x, self.w_qkv)
            attn = self._attention(qkv, kv_cache, layer_idx, pos, b, s,
                                   prefill, native=True).to(self.dtype)
            return gemm_allreduce(attn, self.w_o, self.ar_ctx)
        if self.mode == "allreduce":
            qkv = gemm(x, self.w_qkv)
            attn = self._attention(qkv, kv_cache, layer_idx, pos, b, s,
                                   prefill, native=True).to(self.dtype)
            out = gemm(attn, self.w_o)
            dist.all_reduce(out)
            return out
        return self.torch_fwd(x, kv_cache, layer_idx, pos, b, s, prefill)

    __call__ = forward

    def torch_fwd(self, x, kv_cache=None, layer_idx=0, pos=None, b=1, s=1,
                  prefill=False):
        # bf16 matmul on GPU (hipBLASLt — CDNA4 has no fp32 MFMA, fp32
        # matmul would fall to the 157 TF vector ALU); fp32 on CPU
        if x.is_cuda:
            qkv = x @ self.w_qkv.t()
        else:
            qkv = (x.float() @ self.w_qkv.float().t()).to(self.dtype)
        attn = self._attention(qkv, kv_cache, layer_idx, pos, b, s, prefill)
        if x.is_cuda:
            out = attn.to(self.dtype) @ self.w_o.t()
        else:
            out = (attn.float() @ self.w_o.float().t()).to(self.dtype)
        if dist.is_initialized() and self.world > 1:
            if out.is_cuda and dist.get_backend() == "gloo":
                cpu = out.cpu()
                dist.all_reduce(cpu)
                out = cpu.to(out.device)
            else:
                dist.all_reduce(out)
        return out
