"""Sequence-parallel layer wrappers (reference layer-API parity:
Triton-distributed layers/nvidia/sp_flash_decode_layer.py:44-149,
ulysses_sp_a2a_layer.py:29-49, pre/post_attn_a2a_layer.py — behavior
only; ops in ops/sp.py)."""
from __future__ import annotations

from typing import Optional

import torch

from ..ops.sp import (SPAGAttnContext, SPFlashDecodeContext, UlyssesContext,
                      create_sp_ag_attn_context,
                      create_sp_flash_decode_context, create_ulysses_context,
                      sp_ag_attention, sp_flash_decode, ulysses_a2a)
from ..runtime.symm_mem import SymmHeap, get_heap


class SPFlashDecodeLayer:
    """Sequence-parallel GQA decode: KV sharded by sequence across ranks;
    each rank computes a split-KV partial and the LSE merge combines."""

    def __init__(self, n_heads: int, n_kv_heads: int, head_dim: int = 128,
                 heap: Optional[SymmHeap] = None):
        assert head_dim == 128
        self.heap = heap or get_heap()
        self.qh, self.kvh = n_heads, n_kv_heads
        self.ctx: Optional[SPFlashDecodeContext] = None

    def init_ctx(self, max_batch: int):
        self.ctx = create_sp_flash_decode_context(max_batch, self.qh,
                                                  heap=self.heap)
        return self.ctx

    def forward(self, q: torch.Tensor, kv_k_chunk: torch.Tensor,
                kv_v_chunk: torch.Tensor, chunk_len: torch.Tensor
                ) -> torch.Tensor:
        return sp_flash_decode(q, kv_k_chunk, kv_v_chunk, chunk_len,
                               self.ctx, self.qh, self.kvh)

    __call__ = forward


class UlyssesSPAllToAllLayer:
    """Ulysses head<->sequence resharding around attention: pre-attn
    (tokens sharded, all heads -> all tokens, head shard) and the inverse
    post-attn direction."""

    def __init__(self, n_heads: int, head_dim: int,
                 heap: Optional[SymmHeap] = None):
        self.heap = heap or get_heap()
        self.n_heads, self.head_dim = n_heads, head_dim
        self.pre_ctx: Optional[UlyssesContext] = None
        self.post_ctx: Optional[UlyssesContext] = None

    def init_ctx(self, max_tokens_local: int):
        world = self.heap.world
        self.pre_ctx = create_ulysses_context(max_tokens_local, self.n_heads,
                                              self.head_dim, heap=self.heap)
        # post direction: tokens play the role of heads and vice versa
        self.post_ctx = create_ulysses_context(
            max_tokens_local, self.n_heads, self.head_dim, heap=self.heap)
        return self.pre_ctx

    def pre_attn(self, x: torch.Tensor) -> torch.Tensor:
        """[T_loc, n_heads, D] -> [world*T_loc, n_heads/world, D]."""
        return ulysses_a2a(x, self.pre_ctx)

    def post_attn(self, x: torch.Tensor) -> torch.Tensor:
        """[world*T_loc, h_loc, D] -> [T_loc, n_heads, D] (inverse)."""
        world = self.heap.world
        t_full, h_loc, d = x.shape
        t_loc = t_full // world
        # inverse reshard = forward a2a with (token-chunk <-> head-chunk)
        # roles swapped: view tokens as the "heads" axis
        xi = x.view(world, t_loc, h_loc, d).permute(1, 0, 2, 3) \
            .reshape(t_loc, world * h_loc, d).contiguous()
        out = ulysses_a2a(xi, self.post_ctx)  # [world*t_loc, h_loc, d]
        return out.view(world, t_loc, h_loc, d).permute(1, 0, 2, 3) \
            .reshape(t_loc, world * h_loc, d)

    __call__ = pre_attn


class SPAGAttentionLayer:
    """Ring-AG causal prefill attention (sequence sharded)."""

    def __init__(self, n_heads: int, n_kv_heads: int, head_dim: int,
                 heap: Optional[SymmHeap] = None):
        self.heap = heap or get_heap()
        self.qh, self.kvh, self.d = n_heads, n_kv_heads, head_dim
        self.ctx: Optional[SPAGAttnContext] = None

    def init_ctx(self, max_chunk_tokens: int):
        self.ctx = create_sp_ag_attn_context(max_chunk_tokens, self.kvh,
                                             self.d, heap=self.heap)
        return self.ctx

    def forward(self, q: torch.Tensor, k_chunk: torch.Tensor,
                v_chunk: torch.Tensor) -> torch.Tensor:
        return sp_ag_attention(q, k_chunk, v_chunk, self.ctx, self.qh)

    __call__ = forward
