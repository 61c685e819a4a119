"""RMSNorm + rotary embedding helpers (torch implementations for now; these
are memory-bound rowwise ops that later fuse into the producing HIP kernels).
"""
from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6
             ) -> torch.Tensor:
    dt = x.dtype
    x32 = x.float()
    x32 = x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + eps)
    return (x32 * weight.float()).to(dt)


class Rotary:
    """Precomputed cos/sin tables (host-side trig per the CDNA4 guide —
    on-device sinf/cosf turns RoPE memory-bound into VALU-bound)."""

    def __init__(self, head_dim: int, max_pos: int, base: float = 1e6,
                 device="cpu", dtype=torch.float32):
        inv = 1.0 / (base ** (torch.arange(0, head_dim, 2,
                                           dtype=torch.float32) / head_dim))
        t = torch.arange(max_pos, dtype=torch.float32)
        freqs = torch.outer(t, inv)
        self.cos = freqs.cos().to(device=device, dtype=dtype)
        self.sin = freqs.sin().to(device=device, dtype=dtype)

    def apply(self, q: torch.Tensor, k: torch.Tensor, pos: torch.Tensor):
        """q/k: [B, S, H, D]; pos: [B, S] int64 positions (device tensor —
        graph-capture-safe dynamic indexing)."""
        cos = self.cos[pos].unsqueeze(2)  # [B, S, 1, D/2]
        sin = self.sin[pos].unsqueeze(2)
        return _rope(q, cos, sin), _rope(k, cos, sin)


def _rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
    dt = x.dtype
    x = x.float()
    d2 = x.shape[-1] // 2
    x1, x2 = x[..., :d2], x[..., d2:]
    return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], -1).to(dt)
