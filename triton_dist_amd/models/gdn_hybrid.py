"""Hybrid GDN + attention model (Qwen3-Next geometry): every
`gdn_period`-th layer keeps full attention, the rest use the GDN linear
mixer. The reference ships only the GDN kernels (kernels/nvidia/gdn.py);
this model wiring is an extension in the same direction the reference's
README points (Qwen3-Next support).

Layer loop, Engine, hipGraph capture, TP modes and the fused ag_rs
decode path are all inherited from DenseLLM — GDNMixer is call-
compatible with TP_Attn and its recurrent state lives in HybridCache.
"""
from __future__ import annotations

import torch

from ..layers.gdn_layer import GDNMixer
from .config import ModelConfig
from .dense import DenseLLM
from .kv_cache import HybridCache


class HybridGDNLLM(DenseLLM):
    def __init__(self, cfg: ModelConfig, device="cpu",
                 dtype=torch.bfloat16, heap=None):
        assert cfg.gdn_period > 0, "hybrid model needs gdn_period > 0"
        super().__init__(cfg, device=device, dtype=dtype, heap=heap)
        for li, layer in enumerate(self.layers):
            if (li + 1) % cfg.gdn_period != 0:  # keep every period-th attn
                layer["attn"] = GDNMixer(
                    cfg.hidden, cfg.gdn_heads, cfg.gdn_head_k,
                    cfg.gdn_head_v, mode=self.mode, heap=self.heap,
                    device=device, dtype=dtype)

    def init_weights(self, seed: int = 1234, std: float = 0.02):
        # DenseLLM initializes embed/norms/MLP and the TP_Attn layers; it
        # guards on w_qkv so mixer layers are skipped there
        super().init_weights(seed=seed, std=std)
        for li, layer in enumerate(self.layers):
            attn = layer["attn"]
            if isinstance(attn, GDNMixer):
                attn.init_weights(seed=seed + 9000 + li * 7, std=std)

    def make_cache(self, batch: int, max_len: int) -> HybridCache:
        return HybridCache(self.cfg.n_layers, batch, max_len,
                           max(self.cfg.n_kv_heads // self.world, 1),
                           self.cfg.head_dim, device=self.device,
                           dtype=self.dtype)
