"""Static KV cache (capability parity with Triton-distributed
python/triton_dist/models/kv_cache.py:29-66: [L, B, maxlen, H, D] tensors +
an offset vector). The offset lives on DEVICE and all indexing/masking is
computed from it, so decode steps replay correctly under hipGraph capture.
"""
from __future__ import annotations

import torch


class KVCache:
    def __init__(self, n_layers: int, batch: int, max_len: int,
                 n_kv_heads: int, head_dim: int, device="cpu",
                 dtype=torch.bfloat16):
        self.n_layers, self.batch, self.max_len = n_layers, batch, max_len
        self.kvh, self.head_dim = n_kv_heads, head_dim
        shape = (n_layers, batch, max_len, n_kv_heads, head_dim)
        self.k = torch.zeros(shape, device=device, dtype=dtype)
        self.v = torch.zeros(shape, device=device, dtype=dtype)
        # device-resident so graph replays see the advancing offset
        self.offset = torch.zeros((), device=device, dtype=torch.int64)
        self._range = torch.arange(max_len, device=device)

    def reset(self):
        self.offset.zero_()

    def fill_prefill(self, layer: int, k: torch.Tensor, v: torch.Tensor):
        """k/v: [B, S, kvh, D]; writes positions [0, S). Host-shape static."""
        s = k.shape[1]
        self.k[layer, :, :s].copy_(k)
        self.v[layer, :, :s].copy_(v)

    def append(self, layer: int, k: torch.Tensor, v: torch.Tensor):
        """Decode append at the device offset (graph-safe dynamic index)."""
        s = k.shape[1]
        idx = self.offset + torch.arange(s, device=k.device)
        self.k[layer].index_copy_(1, idx, k)
        self.v[layer].index_copy_(1, idx, v)

    def advance(self, s: int = 1):
        self.offset += s

    def view(self, layer: int, s_new: int):
        """Full-length K/V [B, kvh, maxlen, D] plus a bool mask [1,1,s,maxlen]
        marking valid key positions (<= offset + query index)."""
        ks = self.k[layer].transpose(1, 2)
        vs = self.v[layer].transpose(1, 2)
        qpos = self.offset + torch.arange(s_new, device=ks.device)
        mask = self._range.unsqueeze(0) <= qpos.unsqueeze(1)  # [s, maxlen]
        return ks, vs, mask.view(1, 1, s_new, self.max_len)
