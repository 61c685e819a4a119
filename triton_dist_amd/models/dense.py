"""Dense transformer (Qwen3-32B/8B geometry) assembled from the TP layers.

Capability parity with Triton-distributed python/triton_dist/models/dense.py
(DenseLLM :117-258): per-layer TP_Attn/TP_MLP with shared symmetric contexts
(layer 0 owns, others alias — dense.py:169-208), torch-eager prefill,
triton_dist-mode decode. Weights are random-init (deterministic per seed:
every rank generates the FULL weight and slices its shard, so all TP modes
agree numerically; there is no network for HF checkpoints here).
"""
from __future__ import annotations

from typing import Optional

import torch

from ..layers.norm import Rotary, rms_norm
from ..layers.tp_attn import TP_Attn
from ..layers.tp_mlp import TP_MLP
from ..ops import allgather
from ..runtime.symm_mem import get_heap
from .config import ModelConfig
from .kv_cache import KVCache


class DenseLLM:
    def __init__(self, cfg: ModelConfig, device="cpu",
                 dtype=torch.bfloat16, heap=None):
        self.cfg = cfg
        self.heap = heap or get_heap()
        self.world, self.rank = self.heap.world, self.heap.rank
        self.device, self.dtype = device, dtype
        self.mode = cfg.tp_mode
        self.rotary = Rotary(cfg.head_dim, cfg.max_length, cfg.rope_base,
                             device=device)
        self.embed = torch.empty(cfg.vocab, cfg.hidden, device=device,
                                 dtype=dtype)
        self.lm_head = torch.empty(cfg.vocab, cfg.hidden, device=device,
                                   dtype=dtype)
        self.final_norm_w = torch.ones(cfg.hidden, device=device, dtype=dtype)
        self.layers = []
        for _ in range(cfg.n_layers):
            attn = TP_Attn(cfg.hidden, cfg.n_heads, cfg.n_kv_heads,
                           cfg.head_dim, self.rotary, mode=self.mode,
                           qk_norm=cfg.qk_norm, heap=self.heap, device=device,
                           dtype=dtype, rms_eps=cfg.rms_eps)
            mlp = TP_MLP(cfg.hidden, cfg.intermediate, mode=self.mode,
                         heap=self.heap, device=device, dtype=dtype)
            self.layers.append({
                "attn": attn, "mlp": mlp,
                "ln1": torch.ones(cfg.hidden, device=device, dtype=dtype),
                "ln2": torch.ones(cfg.hidden, device=device, dtype=dtype),
            })
        self._decode_ag_ctx = None  # final-norm allgather before lm_head

    # -------------------------------------------------------------- weights
    def init_weights(self, seed: int = 1234, std: float = 0.02):
        cfg = self.cfg

        def full(shape, s):
            g = torch.Generator(device=self.device).manual_seed(s)
            return (torch.randn(shape, generator=g, device=self.device,
                                dtype=torch.float32) * std).to(self.dtype)

        self.embed.copy_(full(self.embed.shape, seed))
        self.lm_head.copy_(self.embed if cfg.tie_embeddings
                           else full(self.lm_head.shape, seed + 1))
        w, r = self.world, self.rank
        d = cfg.head_dim
        for li, layer in enumerate(self.layers):
            s = seed + 100 + li * 10
            attn, mlp = layer["attn"], layer["mlp"]
            if hasattr(attn, "w_qkv"):  # GDN mixer layers init their own
                # attention-slot weights (models/gdn_hybrid.py); the MLP
                # below is initialized for EVERY layer kind
                # qkv: rows grouped [Q(all heads); K; V] — shard heads
                wq = full((cfg.n_heads * d, cfg.hidden), s)
                wk = full((cfg.n_kv_heads * d, cfg.hidden), s + 1)
                wv = full((cfg.n_kv_heads * d, cfg.hidden), s + 2)
                qh, kvh = attn.qh, attn.kvh
                attn.w_qkv.copy_(torch.cat([
                    wq[r * qh * d:(r + 1) * qh * d],
                    wk[r * kvh * d:(r + 1) * kvh * d],
                    wv[r * kvh * d:(r + 1) * kvh * d]]))
                wo = full((cfg.hidden, cfg.n_heads * d), s + 3)  # K-shard
                attn.w_o.copy_(
                    wo[:, r * qh * d:(r + 1) * qh * d].contiguous())
            if hasattr(mlp, "inter_shard"):  # dense TP_MLP (MoE layers
                # initialize their own expert weights in the subclass)
                i_s = mlp.inter_shard
                wg = full((cfg.intermediate, cfg.hidden), s + 4)
                wu = full((cfg.intermediate, cfg.hidden), s + 5)
                mlp.w_gate_up.copy_(torch.cat([wg[r * i_s:(r + 1) * i_s],
                                               wu[r * i_s:(r + 1) * i_s]]))
                wd = full((cfg.hidden, cfg.intermediate), s + 6)
                mlp.w_down.copy_(wd[:, r * i_s:(r + 1) * i_s].contiguous())

    def make_cache(self, batch: int, max_len: int) -> KVCache:
        """Cache factory (hybrid models override with HybridCache)."""
        return KVCache(self.cfg.n_layers, batch, max_len,
                       max(self.cfg.n_kv_heads // self.world, 1),
                       self.cfg.head_dim, device=self.device,
                       dtype=self.dtype)

    # ------------------------------------------------------------- contexts
    def init_dist_ctx(self, max_m_total: int):
        """Create the shared symmetric contexts for the distributed decode
        path (collective — call on all ranks with the same max_m_total)."""
        if self.mode == "gemm_ar":
            ar0 = None
            for layer in self.layers:
                ar0, _ = layer["attn"].init_ctx(max_m_total, ar0)
                if hasattr(layer["mlp"], "inter_shard"):
                    layer["mlp"].init_ctx(max_m_total, ar0)
            return
        if self.mode != "ag_rs":
            return
        ag0 = rs0 = None
        for layer in self.layers:
            ag0, rs0 = layer["attn"].init_ctx(max_m_total, ag0, rs0)
            if hasattr(layer["mlp"], "inter_shard"):  # dense TP_MLP
                layer["mlp"].init_ctx(max_m_total, ag0, rs0)
        self._decode_ag_ctx = ag0

    # -------------------------------------------------------------- forward
    def step(self, tokens: torch.Tensor, kv: Optional[KVCache],
             pos: torch.Tensor, prefill: bool, mode: Optional[str] = None
             ) -> torch.Tensor:
        """One forward pass. tokens: [B, S] int64; pos: [B, S] positions.
        Returns logits of the LAST position: [B, vocab] (replicated)."""
        mode = mode or self.mode
        b, s = tokens.shape
        m = b * s
        x = self.embed[tokens.reshape(-1)]  # [M, H]
        sharded = (mode == "ag_rs")
        if sharded:
            assert m % self.world == 0
            ms = m // self.world
            x = x[self.rank * ms:(self.rank + 1) * ms].contiguous()
        from ..ops.fused import add_rms_norm_op, rms_norm_op

        eps = self.cfg.rms_eps
        pending = None  # MLP output whose residual add fuses into next norm
        for li, layer in enumerate(self.layers):
            attn, mlp = layer["attn"], layer["mlp"]
            if pending is None:
                h = rms_norm_op(x, layer["ln1"], eps)
            else:
                x, h = add_rms_norm_op(pending, x, layer["ln1"], eps)
            a = attn.forward(h, kv, li, pos, b, s, prefill) \
                if mode == self.mode else \
                attn.torch_fwd(h, kv, li, pos, b, s, prefill)
            x, h = add_rms_norm_op(a, x, layer["ln2"], eps)
            if mode == self.mode:
                pending = mlp.forward(h)
            elif (prefill and h.is_cuda and hasattr(mlp, "prefill_fwd")):
                pending = mlp.prefill_fwd(h)
            else:
                pending = mlp.torch_fwd(h)
        _, x = add_rms_norm_op(pending, x, self.final_norm_w, eps)
        if sharded:
            x = allgather(x, self._decode_ag_ctx)
        # lm_head on the last position of each sequence (replicated compute)
        x_last = x.view(b, s, -1)[:, -1]
        return x_last @ self.lm_head.t()  # [B, vocab] bf16

    def _dist_prefill_ok(self, m_total: int) -> bool:
        """Can the fused ag_rs path run prefill at this token count?
        (Replicated torch prefill all-reduces O(M*H) per layer — at
        TP8/M=512K that is ~690 GB of RCCL traffic; the fused path shards
        it and overlaps, so use it whenever the contexts cover M.)"""
        if self.mode != "ag_rs" or self._decode_ag_ctx is None:
            return False
        ctx = self._decode_ag_ctx
        if m_total % self.world:
            return False
        m = m_total // self.world
        return (m <= ctx.max_m_per_rank and m % 128 == 0
                and m % ctx.chunks_per_rank == 0)

    def prefill(self, tokens: torch.Tensor, kv: KVCache) -> torch.Tensor:
        """Prefill: fused ag_rs path when the contexts cover B*S tokens,
        torch-eager otherwise (cf. reference engine's torch prefill).
        Returns first sampled tokens [B]."""
        b, s = tokens.shape
        pos = torch.arange(s, device=tokens.device).expand(b, s)
        mode = self.mode if self._dist_prefill_ok(b * s) else "torch"
        logits = self.step(tokens, kv, pos, prefill=True, mode=mode)
        kv.advance(s)
        return logits.argmax(-1)

    def decode_step(self, tokens: torch.Tensor, kv: KVCache) -> torch.Tensor:
        """One distributed decode step. tokens: [B] -> next tokens [B].
        Fully device-driven (graph-capturable)."""
        b = tokens.shape[0]
        pos = kv.offset.reshape(1, 1).expand(b, 1)
        logits = self.step(tokens.view(b, 1), kv, pos, prefill=False)
        kv.advance(1)
        return logits.argmax(-1)
