"""Tensor-parallel MLP (SwiGLU) with the reference's three TP modes
(Triton-distributed python/triton_dist/layers/amd/tp_mlp.py:48-145 and mode
switch models/dense.py:84-98 — capability parity, MI355X-native ops):

  ag_rs     — batch-sharded: AG-GEMM(gate|up) -> SwiGLU -> GEMM-RS(down)
  allreduce — replicated activations: local GEMMs -> RCCL all-reduce
  torch     — eager golden reference (RCCL collectives + torch.matmul)
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..ops import (ag_gemm, create_ag_gemm_context, create_allreduce_context,
                   create_gemm_rs_context, gemm, gemm_allreduce, gemm_rs)
from ..runtime.symm_mem import SymmHeap, get_heap


class TP_MLP:
    def __init__(self, hidden: int, intermediate: int, mode: str = "ag_rs",
                 heap: Optional[SymmHeap] = None, device="cpu",
                 dtype=torch.bfloat16):
        self.heap = heap or get_heap()
        self.world, self.rank = self.heap.world, self.heap.rank
        assert intermediate % self.world == 0
        self.hidden = hidden
        self.inter_shard = intermediate // self.world
        self.mode = mode
        self.device, self.dtype = device, dtype
        # fused [gate; up] rows, sharded over ranks; down sharded over K
        self.w_gate_up = torch.empty(2 * self.inter_shard, hidden,
                                     device=device, dtype=dtype)
        self.w_down = torch.empty(hidden, self.inter_shard, device=device,
                                  dtype=dtype)
        self.ag_ctx = None
        self.rs_ctx = None
        self.ar_ctx = None

    def init_weights(self, std=0.02, seed: Optional[int] = None):
        g = None
        if seed is not None:
            g = torch.Generator(device=self.device).manual_seed(seed)
        for w in (self.w_gate_up, self.w_down):
            tmp = torch.randn(w.shape, generator=g, device=self.device,
                              dtype=torch.float32) * std
            w.copy_(tmp.to(self.dtype))

    def init_ctx(self, max_m_total: int, ag_ctx=None, rs_ctx=None):
        """Create (or alias — layer 0 owns, others share, cf. reference
        dense.py:169-208) the symmetric contexts for sequences up to
        max_m_total gathered tokens."""
        if self.mode == "gemm_ar":
            if ag_ctx is None:
                ag_ctx = create_allreduce_context(max_m_total * self.hidden,
                                                  heap=self.heap)
            self.ar_ctx = ag_ctx
            return ag_ctx, None
        if self.mode != "ag_rs":
            return None, None
        assert max_m_total % self.world == 0
        if ag_ctx is None:
            ag_ctx = create_ag_gemm_context(max_m_total // self.world,
                                            self.hidden, heap=self.heap)
        if rs_ctx is None:
            rs_ctx = create_gemm_rs_context(max_m_total, self.hidden,
                                            heap=self.heap)
        self.ag_ctx, self.rs_ctx = ag_ctx, rs_ctx
        return ag_ctx, rs_ctx

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """ag_rs: x is the batch shard [M/world, hidden] -> [M/world, hidden].
        allreduce/torch: x is replicated [M, hidden] -> [M, hidden]."""
        from ..ops.fused import swiglu_op

        if self.mode == "ag_rs":
            h = ag_gemm(x, self.w_gate_up, self.ag_ctx)     # [M, 2*I_s]
            act = swiglu_op(h, self.inter_shard)
            return gemm_rs(act, self.w_down, self.rs_ctx)   # [M/world, hidden]
        if self.mode == "gemm_ar":
            h = gemm(x, self.w_gate_up)
            act = swiglu_op(h, self.inter_shard)
            return gemm_allreduce(act, self.w_down, self.ar_ctx)
        if self.mode == "allreduce":
            h = gemm(x, self.w_gate_up)
            act = swiglu_op(h, self.inter_shard)
            partial = gemm(act, self.w_down)
            dist.all_reduce(partial)
            return partial
        return self.torch_fwd(x)

    __call__ = forward

    def torch_fwd(self, x: torch.Tensor) -> torch.Tensor:
        """Eager golden reference (replicated x [M, hidden]). bf16 matmul on
        GPU (no fp32 MFMA on CDNA4), fp32 on CPU."""
        if x.is_cuda:
            h = x @ self.w_gate_up.t()
            act = (F.silu(h[:, :self.inter_shard].float())
                   * h[:, self.inter_shard:].float()).to(self.dtype)
            out = act @ self.w_down.t()
        else:
            h = x.float() @ self.w_gate_up.float().t()
            act = F.silu(h[:, :self.inter_shard]) * h[:, self.inter_shard:]
            out = (act @ self.w_down.float().t()).to(self.dtype)
        if dist.is_initialized() and self.world > 1:
            if out.is_cuda and dist.get_backend() == "gloo":
                cpu = out.cpu()
                dist.all_reduce(cpu)
                out = cpu.to(out.device)
            else:
                dist.all_reduce(out)
        return out
