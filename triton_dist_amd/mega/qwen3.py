"""Megakernel model builder: one persistent-kernel launch per Qwen3 decode
step (single GPU; multi-GPU megakernel with AR tasks is future work —
the reference's own AMD path had no MoE/mega wiring at all).

Capability parity with Triton-distributed mega_triton_kernel/models/
model_builder.py:86 (ModelBuilder -> tasks) — behavior only.
"""
from __future__ import annotations

import torch

from ..models.dense import DenseLLM
from ..models.kv_cache import KVCache
from .builder import (emit_gemm, MegaGraph, MegaRun, T_ADD_RMSNORM, T_EMBED,
                      T_FLASH_DECODE, T_GEMM_TILE, T_KV_ADVANCE,
                      T_QKV_PROLOGUE, T_RMSNORM, T_SWIGLU)


class MegaQwen3Decode:
    """Builds the decode-step task graph for a DenseLLM once; each step is
    ONE kernel launch (+ a torch argmax for sampling)."""

    def __init__(self, model: DenseLLM, kv: KVCache, batch: int,
                 n_wg: int = 768):
        assert model.world == 1, "megakernel v1 is single-GPU"
        cfg = model.cfg
        assert cfg.head_dim == 128
        self.model, self.kv, self.b = model, kv, batch
        dev = model.device
        bp = (batch + 31) // 32 * 32  # padded rows (32-row GEMM tiles)
        self.bp = bp
        H = cfg.hidden
        qh, kvh, d = cfg.n_heads, cfg.n_kv_heads, cfg.head_dim
        qkv_dim = (qh + 2 * kvh) * d
        i_s = cfg.intermediate

        e = torch.empty
        bf = torch.bfloat16
        self.tokens = torch.zeros(batch, dtype=torch.int64, device=dev)
        self.x = e(bp, H, dtype=bf, device=dev)
        self.h = e(bp, H, dtype=bf, device=dev)
        self.qkv = e(bp, qkv_dim, dtype=bf, device=dev)
        self.q = e(bp, qh * d, dtype=bf, device=dev)
        self.attn = e(bp, qh * d, dtype=bf, device=dev)
        self.attn_o = e(bp, H, dtype=bf, device=dev)
        self.gu = e(bp, 2 * i_s, dtype=bf, device=dev)
        self.act = e(bp, i_s, dtype=bf, device=dev)
        self.mo = e(bp, H, dtype=bf, device=dev)
        self.xn = e(bp, H, dtype=bf, device=dev)
        self.logits = e(bp, cfg.vocab, dtype=bf, device=dev)
        for t in (self.x, self.h, self.qkv, self.q, self.attn, self.attn_o,
                  self.gu, self.act, self.mo, self.xn):
            t.zero_()  # padded rows stay zero

        g = MegaGraph()
        tiles_m = bp // 32

        import os
        # K-split 4 measured best for small-batch decode (14.5 ms vs
        # 20.3 at 1 and 23 at 16 on qwen3-8b bsz 1)
        default_ks = "4" if batch <= 32 else "1"
        ksplit = int(os.environ.get("TD_MK_KSPLIT", default_ks))
        self._ws = {}

        def gemm(a_buf, w, c_buf, n, k, dep):
            # bsz<=4: GEMV route (one task per 512-col chunk; weight-BW
            # bound, ~25x fewer tasks). TD_MK_NO_GEMV=1 forces tiles.
            # GEMV route is opt-in: at 512-col chunks an op spans only
            # ~12 workgroups, collapsing per-op parallelism (measured
            # 137 ms vs 20.6 ms tiles at bsz 1 — per-op latency is
            # parallelism-bound, not task-count-bound)
            if (batch <= 4 and batch * k * 2 <= 40960
                    and os.environ.get("TD_MK_GEMV")):
                from .builder import emit_gemv
                return emit_gemv(g, a_buf.data_ptr(), w.data_ptr(),
                                 c_buf.data_ptr(), batch, n, k, dep)
            ws_ptr = 0
            if ksplit > 1 and k % (64 * ksplit) == 0:
                key = (c_buf.data_ptr(), n)
                if key not in self._ws:
                    self._ws[key] = torch.zeros(ksplit, bp, n,
                                                dtype=torch.float32,
                                                device=dev)
                ws_ptr = self._ws[key].data_ptr()
            return emit_gemm(g, a_buf.data_ptr(), w.data_ptr(),
                             c_buf.data_ptr(), batch, n, k, dep,
                             ksplit=ksplit, ws_ptr=ws_ptr)

        # embed
        emb = g.new_op()
        for r in range(batch):
            g.add_task(T_EMBED, emb, [self.tokens.data_ptr(),
                                      model.embed.data_ptr(),
                                      self.x.data_ptr(), batch, H, r])
        g.next_level()

        pending = None  # (op, delta_buf) awaiting the fused residual add
        prev = emb
        for li, layer in enumerate(model.layers):
            at, ml = layer["attn"], layer["mlp"]
            ln1, ln2 = layer["ln1"], layer["ln2"]
            # h = rms(x + pending?)
            nrm = g.new_op()
            for r in range(batch):
                if pending is None:
                    g.add_task(T_RMSNORM, nrm,
                               [self.x.data_ptr(), ln1.data_ptr(),
                                self.h.data_ptr(), batch, H, r], [(prev, 0)])
                else:
                    g.add_task(T_ADD_RMSNORM, nrm,
                               [self.mo.data_ptr(), self.x.data_ptr(),
                                self.x.data_ptr(), ln1.data_ptr(),
                                self.h.data_ptr(), batch, H, r], [(prev, 0)])
            g.next_level()
            qkv_op = gemm(self.h, at.w_qkv, self.qkv, qkv_dim, H, nrm)
            pro = g.new_op()
            for b in range(batch):
                g.add_task(T_QKV_PROLOGUE, pro,
                           [self.qkv.data_ptr(), self.q.data_ptr(),
                            kv.k[li].data_ptr(), kv.v[li].data_ptr(),
                            model.rotary.cos.data_ptr(),
                            model.rotary.sin.data_ptr(),
                            at.q_norm_w.data_ptr(), at.k_norm_w.data_ptr(),
                            kv.offset.data_ptr(), b, qh, kvh, kv.max_len],
                           [(qkv_op, 0)])
            g.next_level()
            fd = g.new_op()
            for b in range(batch):
                for kh in range(kvh):
                    g.add_task(T_FLASH_DECODE, fd,
                               [self.q.data_ptr(), kv.k[li].data_ptr(),
                                kv.v[li].data_ptr(), self.attn.data_ptr(),
                                kv.offset.data_ptr(), b, kh, qh, kvh,
                                kv.max_len], [(pro, 0)])
            g.next_level()
            o_op = gemm(self.attn, at.w_o, self.attn_o, H, qh * d, fd)
            ar2 = g.new_op()
            for r in range(batch):
                g.add_task(T_ADD_RMSNORM, ar2,
                           [self.attn_o.data_ptr(), self.x.data_ptr(),
                            self.x.data_ptr(), ln2.data_ptr(),
                            self.h.data_ptr(), batch, H, r], [(o_op, 0)])
            g.next_level()
            gu_op = gemm(self.h, ml.w_gate_up, self.gu, 2 * i_s, H, ar2)
            sw = g.new_op()
            nchunks = max(tiles_m * i_s // 1024, 8)
            for c in range(nchunks):
                g.add_task(T_SWIGLU, sw,
                           [self.gu.data_ptr(), self.act.data_ptr(), batch,
                            i_s, c, nchunks], [(gu_op, 0)])
            g.next_level()
            mo_op = gemm(self.act, ml.w_down, self.mo, H, i_s, sw)
            pending, prev = mo_op, mo_op

        # final: xn = rms(x + mo); logits = xn @ lm_head^T; kv.offset += 1
        fin = g.new_op()
        for r in range(batch):
            g.add_task(T_ADD_RMSNORM, fin,
                       [self.mo.data_ptr(), self.x.data_ptr(),
                        self.x.data_ptr(), model.final_norm_w.data_ptr(),
                        self.xn.data_ptr(), batch, H, r], [(prev, 0)])
        g.next_level()
        vocab_pad = cfg.vocab // 128 * 128
        lm = gemm(self.xn, model.lm_head, self.logits, vocab_pad, H, fin)
        adv = g.new_op()
        g.add_task(T_KV_ADVANCE, adv, [kv.offset.data_ptr()], [(lm, 0)])
        g.next_level()
        self.run = MegaRun(g, n_wg=n_wg, device=dev)
        self.n_tasks = len(g.tasks)
        self.vocab_pad = vocab_pad

    def step(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens [B] -> logits [B, vocab_pad] (one megakernel launch)."""
        self.tokens.copy_(tokens)
        self.run.launch()
        return self.logits[:self.b, :self.vocab_pad]

    def decode_once(self, tokens: torch.Tensor) -> torch.Tensor:
        return self.step(tokens).argmax(-1)
