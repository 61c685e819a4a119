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
                      T_FLASH_DECODE, T_GEMM_TILE, T_GEMM_TILE_PART,
                      T_PRO_FLASH_DECODE, T_TILE_REDUCE, T_KV_ADVANCE,
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

        # Tile-granular pipelined MLP chain (TD_MK_PIPE=1, bsz=1): the
        # whole-op level-synchronized deps measured 70% WG wait
        # (profiles/README.md r02). Here the chain is split into G
        # column groups with K-sliced deps — gate_up tiles of group g
        # feed ONLY group-g swiglu chunks, which feed ONLY the down
        # K-slice s=g — and the emission levels interleave group chains,
        # so each workgroup's queue is a software pipeline (group g+1
        # parts run while group g's tail still drains).
        GPIPE = 4
        pipe = (os.environ.get("TD_MK_PIPE") == "1" and batch == 1
                and ksplit in (1, GPIPE) and i_s % (GPIPE * 128) == 0
                and (2 * i_s) % 128 == 0 and H % 128 == 0)

        # Hop fusion (TD_MK_FUSE=1): the three 1-task-per-row
        # serialization hops per layer — [add+]rmsnorm before qkv,
        # qkv-prologue, [add+]rmsnorm before gate_up — fold into their
        # consumers (norm into the GEMM K-partials' A staging, prologue
        # into each (b, kh) flash-decode task). The residual UPDATE runs
        # as a parallel task writing a PING-PONG x buffer (readers of
        # (x, delta) never see a half-updated row); TD_MK_PROF measured
        # 70% WG wait, and these hops are the per-layer critical path.
        fuse = (os.environ.get("TD_MK_FUSE", "0") == "1" and batch <= 32
                and ksplit > 1 and H % (64 * ksplit) == 0)
        self.fused = fuse
        if fuse:
            self.x2 = e(bp, H, dtype=bf, device=dev)
            self.x2.zero_()

        def gemm_nr(x_t, lnw_t, res_ptr, w, c_buf, n, k, dep, dep2):
            key = (c_buf.data_ptr(), n)
            if key not in self._ws:
                self._ws[key] = torch.zeros(ksplit, bp, n,
                                            dtype=torch.float32,
                                            device=dev)
            return emit_gemm(g, x_t.data_ptr(), w.data_ptr(),
                             c_buf.data_ptr(), batch, n, k, dep,
                             ksplit=ksplit,
                             ws_ptr=self._ws[key].data_ptr(),
                             norm=(lnw_t.data_ptr(), res_ptr), dep2=dep2)

        def emit_mlp_pipe(ml, dep):
            I = i_s
            gu_ptr, act_ptr, mo_ptr = (self.gu.data_ptr(),
                                       self.act.data_ptr(),
                                       self.mo.data_ptr())
            # fp32 ws for both GEMMs' K-split parts
            for key_ptr, n_out in ((gu_ptr, 2 * I), (mo_ptr, H)):
                kkey = (key_ptr, n_out)
                if kkey not in self._ws:
                    self._ws[kkey] = torch.zeros(GPIPE, bp, n_out,
                                                 dtype=torch.float32,
                                                 device=dev)
            ws_gu = self._ws[(gu_ptr, 2 * I)].data_ptr()
            ws_mo = self._ws[(mo_ptr, H)].data_ptr()
            h_ptr = self.h.data_ptr()
            wgu = ml.w_gate_up.data_ptr()
            wdn = ml.w_down.data_ptr()
            tiles_gu = (2 * I) // 128
            tiles_dn = H // 128
            kgu = H          # gate_up K
            klen_gu = kgu // GPIPE
            klen_dn = I // GPIPE
            nchunks = GPIPE * 4     # swiglu chunks (4 per group)
            cpg = nchunks // GPIPE
            RG = [g.new_op() for _ in range(GPIPE)]
            SW = [g.new_op() for _ in range(GPIPE)]
            DPS = [g.new_op() for _ in range(tiles_dn)]
            base_lvl = g.level
            # gate_up: per-tile part ops + group reduces
            for tn in range(tiles_gu):
                n0 = tn * 128
                pc = n0 if n0 < I else n0 - I
                grp = pc * GPIPE // I
                g.set_level(base_lvl + grp * 5)
                pt = g.new_op()
                for sk2 in range(GPIPE):
                    g.add_task(T_GEMM_TILE_PART, pt,
                               [h_ptr, wgu, ws_gu, batch, 2 * I, kgu, 0,
                                tn, sk2 * klen_gu, klen_gu, sk2],
                               [(dep, 0)])
                g.set_level(base_lvl + grp * 5 + 1)
                g.add_task(T_TILE_REDUCE, RG[grp],
                           [ws_gu, gu_ptr, batch, 2 * I, 0, tn, GPIPE],
                           [(pt, 0)])
            # swiglu per group
            for grp in range(GPIPE):
                g.set_level(base_lvl + grp * 5 + 2)
                for c in range(grp * cpg, (grp + 1) * cpg):
                    g.add_task(T_SWIGLU, SW[grp],
                               [gu_ptr, act_ptr, batch, I, c, nchunks],
                               [(RG[grp], 0)])
            # down: per-tile K-slice parts (slice s deps only group s)
            for grp in range(GPIPE):
                g.set_level(base_lvl + grp * 5 + 3)
                for dt in range(tiles_dn):
                    g.add_task(T_GEMM_TILE_PART, DPS[dt],
                               [act_ptr, wdn, ws_mo, batch, H, I, 0, dt,
                                grp * klen_dn, klen_dn, grp],
                               [(SW[grp], 0)])
            g.set_level(base_lvl + (GPIPE - 1) * 5 + 4)
            dr = g.new_op()
            for dt in range(tiles_dn):
                g.add_task(T_TILE_REDUCE, dr,
                           [ws_mo, mo_ptr, batch, H, 0, dt, GPIPE],
                           [(DPS[dt], 0)])
            g.set_level(base_lvl + (GPIPE - 1) * 5 + 5)
            return dr

        # embed
        emb = g.new_op()
        for r in range(batch):
            g.add_task(T_EMBED, emb, [self.tokens.data_ptr(),
                                      model.embed.data_ptr(),
                                      self.x.data_ptr(), batch, H, r])
        g.next_level()

        pending = None  # (op, delta_buf) awaiting the fused residual add
        prev = emb
        xup_prev = None            # ping-pong x-update op (fused mode)
        x_cur, x_alt = self.x, (self.x2 if fuse else self.x)
        for li, layer in enumerate(model.layers):
            at, ml = layer["attn"], layer["mlp"]
            ln1, ln2 = layer["ln1"], layer["ln2"]
            if fuse:
                res_ptr = self.mo.data_ptr() if pending is not None else 0
                if pending is not None:
                    # parallel (off critical path): x_alt = x_cur + mo
                    xup = g.new_op()
                    for r in range(batch):
                        g.add_task(
                            T_ADD_RMSNORM, xup,
                            [self.mo.data_ptr(), x_cur.data_ptr(),
                             x_alt.data_ptr(), ln1.data_ptr(),
                             self.h.data_ptr(), batch, H, r],
                            [(prev, 0)] + ([(xup_prev, 0)] if xup_prev
                                           else []))
                qkv_op = gemm_nr(x_cur, ln1, res_ptr, at.w_qkv, self.qkv,
                                 qkv_dim, H, prev, xup_prev)
                if pending is not None:
                    xup_prev = xup
                    x_cur, x_alt = x_alt, x_cur
            else:
                # h = rms(x + pending?)
                nrm = g.new_op()
                for r in range(batch):
                    if pending is None:
                        g.add_task(T_RMSNORM, nrm,
                                   [self.x.data_ptr(), ln1.data_ptr(),
                                    self.h.data_ptr(), batch, H, r],
                                   [(prev, 0)])
                    else:
                        g.add_task(T_ADD_RMSNORM, nrm,
                                   [self.mo.data_ptr(), self.x.data_ptr(),
                                    self.x.data_ptr(), ln1.data_ptr(),
                                    self.h.data_ptr(), batch, H, r],
                                   [(prev, 0)])
                g.next_level()
                qkv_op = gemm(self.h, at.w_qkv, self.qkv, qkv_dim, H, nrm)
            if fuse:
                # prologue folded into each (b, kh) flash-decode task
                fd = g.new_op()
                for b in range(batch):
                    for kh in range(kvh):
                        g.add_task(
                            T_PRO_FLASH_DECODE, fd,
                            [self.qkv.data_ptr(), self.q.data_ptr(),
                             kv.k[li].data_ptr(), kv.v[li].data_ptr(),
                             model.rotary.cos.data_ptr(),
                             model.rotary.sin.data_ptr(),
                             at.q_norm_w.data_ptr(),
                             at.k_norm_w.data_ptr(),
                             kv.offset.data_ptr(), self.attn.data_ptr(),
                             (kh << 32) | b, (kvh << 32) | qh,
                             kv.max_len], [(qkv_op, 0)])
                g.next_level()
            else:
                pro = g.new_op()
                for b in range(batch):
                    g.add_task(T_QKV_PROLOGUE, pro,
                               [self.qkv.data_ptr(), self.q.data_ptr(),
                                kv.k[li].data_ptr(), kv.v[li].data_ptr(),
                                model.rotary.cos.data_ptr(),
                                model.rotary.sin.data_ptr(),
                                at.q_norm_w.data_ptr(),
                                at.k_norm_w.data_ptr(),
                                kv.offset.data_ptr(), b, qh, kvh,
                                kv.max_len],
                               [(qkv_op, 0)])
                g.next_level()
                fd = g.new_op()
                for b in range(batch):
                    for kh in range(kvh):
                        g.add_task(T_FLASH_DECODE, fd,
                                   [self.q.data_ptr(), kv.k[li].data_ptr(),
                                    kv.v[li].data_ptr(),
                                    self.attn.data_ptr(),
                                    kv.offset.data_ptr(), b, kh, qh, kvh,
                                    kv.max_len], [(pro, 0)])
                g.next_level()
            o_op = gemm(self.attn, at.w_o, self.attn_o, H, qh * d, fd)
            if fuse:
                xup2 = g.new_op()  # parallel: x_alt = x_cur + attn_o
                for r in range(batch):
                    g.add_task(
                        T_ADD_RMSNORM, xup2,
                        [self.attn_o.data_ptr(), x_cur.data_ptr(),
                         x_alt.data_ptr(), ln2.data_ptr(),
                         self.h.data_ptr(), batch, H, r],
                        [(o_op, 0)] + ([(xup_prev, 0)] if xup_prev
                                       else []))
                gu_op = gemm_nr(x_cur, ln2, self.attn_o.data_ptr(),
                                ml.w_gate_up, self.gu, 2 * i_s, H, o_op,
                                xup_prev)
                xup_prev = xup2
                x_cur, x_alt = x_alt, x_cur
                sw = g.new_op()
                nchunks = max(tiles_m * i_s // 1024, 8)
                for c in range(nchunks):
                    g.add_task(T_SWIGLU, sw,
                               [self.gu.data_ptr(), self.act.data_ptr(),
                                batch, i_s, c, nchunks], [(gu_op, 0)])
                g.next_level()
                mo_op = gemm(self.act, ml.w_down, self.mo, H, i_s, sw)
            else:
                ar2 = g.new_op()
                for r in range(batch):
                    g.add_task(T_ADD_RMSNORM, ar2,
                               [self.attn_o.data_ptr(), self.x.data_ptr(),
                                self.x.data_ptr(), ln2.data_ptr(),
                                self.h.data_ptr(), batch, H, r],
                               [(o_op, 0)])
                g.next_level()
                if pipe:
                    mo_op = emit_mlp_pipe(ml, ar2)
                else:
                    gu_op = gemm(self.h, ml.w_gate_up, self.gu, 2 * i_s,
                                 H, ar2)
                    sw = g.new_op()
                    nchunks = max(tiles_m * i_s // 1024, 8)
                    for c in range(nchunks):
                        g.add_task(T_SWIGLU, sw,
                                   [self.gu.data_ptr(),
                                    self.act.data_ptr(),
                                    batch, i_s, c, nchunks], [(gu_op, 0)])
                    g.next_level()
                    mo_op = gemm(self.act, ml.w_down, self.mo, H, i_s, sw)
            pending, prev = mo_op, mo_op

        # final: xn = rms(x + mo); logits = xn @ lm_head^T; kv.offset += 1
        fin = g.new_op()
        for r in range(batch):
            g.add_task(T_ADD_RMSNORM, fin,
                       [self.mo.data_ptr(), x_cur.data_ptr(),
                        x_alt.data_ptr(), model.final_norm_w.data_ptr(),
                        self.xn.data_ptr(), batch, H, r],
                       [(prev, 0)] + ([(xup_prev, 0)] if xup_prev else []))
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
