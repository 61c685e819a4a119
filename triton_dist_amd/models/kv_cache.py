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


class PagedKVCache:
    """Block-paged KV pool (capability parity with the reference
    megakernel's paged cache, mega_triton_kernel/models/paged_kv_cache.py
    — behavior only): a shared pool of fixed-size blocks
    [n_blocks, block, kvh, D] per side with a per-sequence block table,
    blocks allocated on first touch. Serving engines get allocation at
    block granularity instead of max_len * batch up front.

    The HIP decode kernels consume contiguous [B, L, kvh, D];
    `gather_layer` materializes that view from the table (a functional
    bridge — a natively paged flash-decode indexing the table in-kernel
    is roadmap)."""

    def __init__(self, n_layers: int, batch: int, max_len: int,
                 n_kv_heads: int, head_dim: int, block: int = 64,
                 n_blocks: int = 0, device="cpu", dtype=torch.bfloat16):
        assert max_len % block == 0
        self.n_layers, self.batch = n_layers, batch
        self.max_len, self.block = max_len, block
        self.kvh, self.head_dim = n_kv_heads, head_dim
        self.blocks_per_seq = max_len // block
        if n_blocks <= 0:
            n_blocks = batch * self.blocks_per_seq  # worst case
        self.n_blocks = n_blocks
        pool = (n_layers, n_blocks, block, n_kv_heads, head_dim)
        self.k_pool = torch.zeros(pool, device=device, dtype=dtype)
        self.v_pool = torch.zeros(pool, device=device, dtype=dtype)
        # block_table[b, j] = pool block id of sequence b's j-th block
        # (-1 = unallocated); shared across layers (same shape every layer)
        self.block_table = torch.full((batch, self.blocks_per_seq), -1,
                                      dtype=torch.int64, device=device)
        self.seq_len = torch.zeros(batch, dtype=torch.int64, device=device)
        self._free_top = 0

    def reset(self):
        self.block_table.fill_(-1)
        self.seq_len.zero_()
        self._free_top = 0

    def _ensure_blocks(self, upto_len: int):
        """Allocate pool blocks for every sequence up to `upto_len`
        positions (host-side allocator; block ids land in the table)."""
        need = (upto_len + self.block - 1) // self.block
        table = self.block_table
        for j in range(need):
            col = table[:, j]
            missing = (col < 0).nonzero(as_tuple=True)[0]
            if missing.numel():
                n = missing.numel()
                if self._free_top + n > self.n_blocks:
                    raise RuntimeError("PagedKVCache: pool exhausted")
                ids = torch.arange(self._free_top, self._free_top + n,
                                   device=table.device)
                table[missing, j] = ids
                self._free_top += n

    def append(self, layer: int, k: torch.Tensor, v: torch.Tensor,
               pos0: int):
        """k/v: [B, S, kvh, D] written at positions [pos0, pos0+S)."""
        b, s = k.shape[0], k.shape[1]
        self._ensure_blocks(pos0 + s)
        for off in range(s):
            p = pos0 + off
            blk = self.block_table[:, p // self.block]     # [B]
            self.k_pool[layer, blk, p % self.block] = k[:, off]
            self.v_pool[layer, blk, p % self.block] = v[:, off]
        if layer == self.n_layers - 1:
            self.seq_len.fill_(pos0 + s)

    def gather_layer(self, layer: int, upto_len: int):
        """Contiguous [B, upto_len, kvh, D] views for the decode kernels."""
        nb = (upto_len + self.block - 1) // self.block
        blk = self.block_table[:, :nb].clamp(min=0)        # [B, nb]
        k = self.k_pool[layer, blk]                        # [B, nb, blk,...]
        v = self.v_pool[layer, blk]
        shp = (self.batch, nb * self.block, self.kvh, self.head_dim)
        return k.reshape(shp)[:, :upto_len], v.reshape(shp)[:, :upto_len]


class HybridCache(KVCache):
    """KVCache + per-GDN-layer recurrent state for hybrid (Qwen3-Next
    style) stacks. GDN state is positionless fp32 [B, local_heads, K, V],
    zeroed on reset; attention layers use the inherited paged-less KV
    slots (GDN layers simply never touch theirs)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._gdn = {}

    def gdn_state(self, layer: int, batch: int, lh: int, dk: int,
                  dv: int) -> "torch.Tensor":
        key = layer
        if key not in self._gdn:
            self._gdn[key] = torch.zeros(batch, lh, dk, dv,
                                         dtype=torch.float32,
                                         device=self.k.device)
        return self._gdn[key]

    def reset(self):
        super().reset()
        for t in self._gdn.values():
            t.zero_()
