"""GDN mixer layer — the linear-attention block of a Qwen3-Next-style
hybrid stack. Head-sharded TP like TP_Attn (each rank owns
gdn_heads/world heads and the matching out-proj columns).

The reference ships GDN kernels only (kernels/nvidia/gdn.py); the model
wiring here goes beyond it: a drop-in mixer with TP_Attn's call
signature so DenseLLM's layer loop drives either. Decode keeps the fp32
[K,V] state per (batch, local head) in-place on device (hipGraph-safe:
no position cell — linear attention state is positionless); prefill
runs the chunked WY form and installs the final state.

Simplifications vs HF Qwen3-Next (documented, this is the GDN-geometry
family, not a checkpoint-compatible port): no short conv, gate/beta are
direct per-head projections (g = logsigmoid, beta = sigmoid).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ..ops.gdn import chunk_gated_delta_rule_fwd, gdn_decode_step
from ..runtime.symm_mem import SymmHeap, get_heap


class GDNMixer:
    def __init__(self, hidden: int, gdn_heads: int, head_k: int,
                 head_v: int, mode: str = "ag_rs",
                 heap: Optional[SymmHeap] = None, device="cpu",
                 dtype=torch.bfloat16):
        self.heap = heap or get_heap()
        self.world, self.rank = self.heap.world, self.heap.rank
        assert gdn_heads % self.world == 0
        self.hidden = hidden
        self.heads = gdn_heads
        self.lh = gdn_heads // self.world  # local heads
        self.dk, self.dv = head_k, head_v
        self.mode = mode
        self.device, self.dtype = device, dtype
        # fused qkvgb projection: rows [q(lh*dk); k(lh*dk); v(lh*dv);
        # g(lh); beta(lh); zero-pad] — one GEMM per step like TP_Attn's
        # w_qkv. Padded to a multiple of 128 so the fused AG-GEMM
        # consumers' N-tiling constraint holds for any head count.
        used = self.lh * (2 * self.dk + self.dv + 2)
        self.proj_dim = (used + 127) & ~127
        self._proj_used = used
        self.w_in = torch.empty(self.proj_dim, hidden, device=device,
                                dtype=dtype)
        self.w_out = torch.empty(hidden, self.lh * self.dv, device=device,
                                 dtype=dtype)
        self.scale = head_k ** -0.5
        self.ag_ctx = None
        self.rs_ctx = None

    # ------------------------------------------------------------ weights
    def init_weights(self, seed: int, std: float = 0.02):
        g = torch.Generator(device=self.device).manual_seed(seed)

        def full(shape):
            return (torch.randn(shape, generator=g, device=self.device,
                                dtype=torch.float32) * std).to(self.dtype)

        # generate FULL (unsharded) then slice this rank's heads so every
        # world size sees the same model
        dk, dv, H = self.dk, self.dv, self.heads
        wq = full((H * dk, self.hidden))
        wk = full((H * dk, self.hidden))
        wv = full((H * dv, self.hidden))
        wg = full((H, self.hidden))
        wb = full((H, self.hidden))
        wo = full((self.hidden, H * dv))
        r, lh = self.rank, self.lh
        pad = torch.zeros(self.proj_dim - self._proj_used, self.hidden,
                          device=self.device, dtype=self.dtype)
        self.w_in.copy_(torch.cat([
            wq[r * lh * dk:(r + 1) * lh * dk],
            wk[r * lh * dk:(r + 1) * lh * dk],
            wv[r * lh * dv:(r + 1) * lh * dv],
            wg[r * lh:(r + 1) * lh],
            wb[r * lh:(r + 1) * lh],
            pad]))
        self.w_out.copy_(wo[:, r * lh * dv:(r + 1) * lh * dv].contiguous())

    # ----------------------------------------------------------- contexts
    def init_ctx(self, max_m_total: int, ag_ctx=None, rs_ctx=None):
        """Same contract as TP_Attn.init_ctx: returns the shared (ag, rs)
        pair so DenseLLM's chain threads through mixer layers too."""
        if self.mode == "gemm_ar":
            from ..ops.allreduce import create_allreduce_context

            if ag_ctx is None:
                ag_ctx = create_allreduce_context(
                    max_m_total * self.hidden, heap=self.heap)
            self.ar_ctx = ag_ctx
            return ag_ctx, None
        if self.mode != "ag_rs":
            return ag_ctx, rs_ctx
        from ..ops.allgather_gemm import create_ag_gemm_context
        from ..ops.gemm_rs import create_gemm_rs_context

        if ag_ctx is None:
            ag_ctx = create_ag_gemm_context(max_m_total // self.world,
                                            self.hidden, heap=self.heap)
        if rs_ctx is None:
            rs_ctx = create_gemm_rs_context(max_m_total, self.hidden,
                                            heap=self.heap)
        self.ag_ctx, self.rs_ctx = ag_ctx, rs_ctx
        return ag_ctx, rs_ctx

    # ------------------------------------------------------------ forward
    def _split(self, proj: torch.Tensor, m: int):
        lh, dk, dv = self.lh, self.dk, self.dv
        q = proj[:, :lh * dk].reshape(m, lh, dk)
        k = proj[:, lh * dk:2 * lh * dk].reshape(m, lh, dk)
        v = proj[:, 2 * lh * dk:2 * lh * dk + lh * dv].reshape(m, lh, dv)
        gb = proj[:, 2 * lh * dk + lh * dv:self._proj_used].float()
        g = F.logsigmoid(gb[:, :lh])
        beta = torch.sigmoid(gb[:, lh:])
        k = F.normalize(k.float(), p=2, dim=-1).to(k.dtype)
        return q, k, v, g, beta

    def forward(self, x: torch.Tensor, kv_cache=None, layer_idx: int = 0,
                pos: Optional[torch.Tensor] = None, b: int = 1, s: int = 1,
                prefill: bool = False) -> torch.Tensor:
        """TP_Attn-compatible signature; kv_cache/pos unused (state is
        internal and positionless)."""
        from ..ops.allgather_gemm import ag_gemm
        from ..ops.gemm_rs import gemm_rs
        from ..ops.gemm import best_gemm

        if self.mode == "ag_rs":
            proj = ag_gemm(x, self.w_in, self.ag_ctx)
        else:
            proj = x.to(self.dtype) @ self.w_in.t()
        m = proj.shape[0]
        q, k, v, g, beta = self._split(proj, m)

        state = kv_cache.gdn_state(layer_idx, b, self.lh, self.dk,
                                   self.dv)
        if prefill:
            # [B, S, lh, *] chunked forward; final state installed
            qb = q.reshape(b, s, self.lh, self.dk)
            kb = k.reshape(b, s, self.lh, self.dk)
            vb = v.reshape(b, s, self.lh, self.dv)
            gb_ = g.reshape(b, s, self.lh)
            bb = beta.reshape(b, s, self.lh)
            o, final = chunk_gated_delta_rule_fwd(qb, kb, vb, gb_, bb,
                                                  self.scale)
            state.copy_(final)
            o = o.reshape(m, self.lh * self.dv).to(self.dtype)
        else:
            assert s == 1
            o = gdn_decode_step(q, k, v, g, beta, self.scale, state)
            o = o.reshape(m, self.lh * self.dv).to(self.dtype)

        if self.mode == "ag_rs":
            return gemm_rs(o, self.w_out, self.rs_ctx)
        if self.mode == "gemm_ar" and getattr(self, "ar_ctx", None) \
                is not None:
            from ..ops.allreduce import gemm_allreduce

            return gemm_allreduce(o, self.w_out, self.ar_ctx)
        partial = best_gemm(o, self.w_out) if o.is_cuda \
            else (o.float() @ self.w_out.float().t()).to(self.dtype)
        if self.mode in ("allreduce", "gemm_ar") and self.world > 1:
            import torch.distributed as dist
            partial = partial.float()
            dist.all_reduce(partial)
            partial = partial.to(self.dtype)
        return partial

    __call__ = forward

    def torch_fwd(self, x, kv_cache=None, layer_idx=0, pos=None, b=1, s=1,
                  prefill=False):
        """Replicated-input golden path: local-head GDN + out-proj partial
        + all-reduce (mirrors TP_Attn.torch_fwd's contract)."""
        import torch.distributed as dist

        if x.is_cuda:
            proj = x @ self.w_in.t()
        else:
            proj = (x.float() @ self.w_in.float().t()).to(self.dtype)
        m = proj.shape[0]
        q, k, v, g, beta = self._split(proj, m)
        state = kv_cache.gdn_state(layer_idx, b, self.lh, self.dk, self.dv)
        if prefill:
            o, final = chunk_gated_delta_rule_fwd(
                q.reshape(b, s, self.lh, self.dk),
                k.reshape(b, s, self.lh, self.dk),
                v.reshape(b, s, self.lh, self.dv),
                g.reshape(b, s, self.lh), beta.reshape(b, s, self.lh),
                self.scale)
            state.copy_(final)
        else:
            o = gdn_decode_step(q, k, v, g, beta, self.scale, state)
        o = o.reshape(m, self.lh * self.dv).to(self.dtype)
        if x.is_cuda:
            out = o @ self.w_out.t()
        else:
            out = (o.float() @ self.w_out.float().t()).to(self.dtype)
        if dist.is_initialized() and self.world > 1:
            if out.is_cuda and dist.get_backend() == "gloo":
                cpu = out.cpu()
                dist.all_reduce(cpu)
                out = cpu.to(out.device)
            else:
                out = out.float()
                dist.all_reduce(out)
                out = out.to(self.dtype)
        return out
